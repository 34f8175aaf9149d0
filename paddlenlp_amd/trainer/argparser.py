"""PdArgumentParser: dataclass -> argparse with JSON-config + CLI overrides.

Reference: paddlenlp/trainer/argparser.py (parse_json_file_and_cmd_lines is
what every llm/ entry script uses, e.g. llm/run_pretrain.py:362).
"""
from __future__ import annotations

import dataclasses
import json
import sys
from argparse import ArgumentParser
from enum import Enum
from pathlib import Path
from typing import List, Tuple, Union, get_args, get_origin


def _str2bool(v):
    if isinstance(v, bool):
        return v
    if v.lower() in ("yes", "true", "t", "1"):
        return True
    if v.lower() in ("no", "false", "f", "0"):
        return False
    raise ValueError(f"Not a bool: {v}")


class PdArgumentParser(ArgumentParser):
    def __init__(self, dataclass_types, **kwargs):
        super().__init__(**kwargs)
        if dataclasses.is_dataclass(dataclass_types):
            dataclass_types = [dataclass_types]
        self.dataclass_types = list(dataclass_types)
        for dtype in self.dataclass_types:
            self._add_dataclass_arguments(dtype)

    def _add_dataclass_arguments(self, dtype):
        for f in dataclasses.fields(dtype):
            if not f.init:
                continue
            name = f"--{f.name}"
            kwargs: dict = {}
            ftype = f.type
            if isinstance(ftype, str):
                # from __future__ annotations: resolve basic names
                ftype = {"str": str, "int": int, "float": float, "bool": bool}.get(
                    ftype.replace("Optional[", "").rstrip("]"), str
                )
            origin = get_origin(ftype)
            if origin is Union:
                non_none = [a for a in get_args(ftype) if a is not type(None)]
                ftype = non_none[0] if non_none else str
                origin = get_origin(ftype)
            if origin in (list, List):
                kwargs["nargs"] = "+"
                args_ = get_args(ftype)
                kwargs["type"] = args_[0] if args_ else str
            elif ftype is bool:
                kwargs["type"] = _str2bool
                kwargs["nargs"] = "?"
                kwargs["const"] = True
            elif isinstance(ftype, type) and issubclass(ftype, Enum):
                kwargs["type"] = str
                kwargs["choices"] = [e.value for e in ftype]
            else:
                kwargs["type"] = ftype if callable(ftype) else str
            if f.default is not dataclasses.MISSING:
                kwargs["default"] = f.default
            elif f.default_factory is not dataclasses.MISSING:
                kwargs["default"] = f.default_factory()
            else:
                kwargs["required"] = True
            self.add_argument(name, **kwargs)

    def _build(self, namespace) -> Tuple:
        outputs = []
        remaining = dict(vars(namespace))
        for dtype in self.dataclass_types:
            keys = {f.name for f in dataclasses.fields(dtype) if f.init}
            inputs = {k: remaining.pop(k) for k in list(remaining) if k in keys}
            outputs.append(dtype(**inputs))
        return tuple(outputs)

    def parse_args_into_dataclasses(self, args=None, return_remaining_strings=False):
        namespace, remaining = self.parse_known_args(args=args)
        outputs = self._build(namespace)
        if return_remaining_strings:
            return (*outputs, remaining)
        if remaining:
            raise ValueError(f"Unknown arguments: {remaining}")
        return outputs

    def parse_json_file(self, json_file: str):
        data = json.loads(Path(json_file).read_text())
        return self.parse_dict(data)

    def parse_dict(self, data: dict):
        argv = []
        for k, v in data.items():
            argv.append(f"--{k}")
            if isinstance(v, bool):
                argv.append(str(v))
            elif isinstance(v, list):
                argv.extend(str(x) for x in v)
            else:
                argv.append(str(v))
        return self.parse_args_into_dataclasses(args=argv)

    def parse_json_file_and_cmd_lines(self, args=None):
        """argv[1] is a JSON config; later CLI flags override its values."""
        if args is None:
            args = sys.argv[1:]
        if args and args[0].endswith(".json"):
            data = json.loads(Path(args[0]).read_text())
            cli = args[1:]
        else:
            data, cli = {}, list(args)
        argv = []
        # JSON first, CLI after (argparse keeps the last occurrence)
        for k, v in data.items():
            argv.append(f"--{k}")
            if isinstance(v, bool):
                argv.append(str(v))
            elif isinstance(v, list):
                argv.extend(str(x) for x in v)
            else:
                argv.append(str(v))
        argv.extend(cli)
        return self.parse_args_into_dataclasses(args=argv)
