"""Every config JSON shipped by the reference must parse through this
framework's entry-point argument classes (a reference user's config files
work unchanged).  Presets that declare tp/pp/sharding worlds >1 correctly
fail only the world-size validation when parsed in a single process.
"""
import glob
import os
import re
import sys

import pytest

REF = "/root/reference/llm/config"

pytestmark = pytest.mark.skipif(not os.path.isdir(REF),
                                reason="reference tree not present")


def _parsers():
    sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "llm"))
    from paddlenlp_amd.trainer import PdArgumentParser
    from run_finetune import DataArgument, FinetuneArguments, ModelArgument
    from run_pretrain import (
        DataArguments,
        ModelArguments,
        PreTrainingArguments,
    )
    import importlib.util

    spec = importlib.util.spec_from_file_location(
        "run_dpo_mod",
        os.path.join(os.path.dirname(__file__), "..", "llm",
                     "alignment", "dpo", "run_dpo.py"))
    rd = importlib.util.module_from_spec(spec)
    sys.modules["run_dpo_mod"] = rd
    spec.loader.exec_module(rd)
    return {
        "ft": PdArgumentParser((ModelArgument, DataArgument,
                                FinetuneArguments)),
        "pt": PdArgumentParser((ModelArguments, DataArguments,
                                PreTrainingArguments)),
        "dp": PdArgumentParser((rd.ModelArgument, rd.DataArgument,
                                rd.DPOArguments)),
    }


def test_all_reference_presets_schema_compatible():
    saved_argv = sys.argv
    sys.argv = ["x"]
    try:
        parsers = _parsers()
        bad = []
        for f in sorted(glob.glob(os.path.join(REF, "*", "*.json"))):
            name = os.path.basename(f)
            if "pretrain" in name:
                parser = parsers["pt"]
            elif any(k in name for k in ("dpo", "kto", "simpo", "orpo")):
                parser = parsers["dp"]
            else:
                parser = parsers["ft"]
            try:
                parser.parse_json_file(f)
            except Exception as e:
                # multi-GPU presets validly reject a world of 1
                if "not divisible" not in str(e):
                    bad.append((name, str(e)[:90]))
        assert not bad, bad
    finally:
        sys.argv = saved_argv
