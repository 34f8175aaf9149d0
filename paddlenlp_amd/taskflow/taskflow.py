"""Taskflow: task-name -> pipeline dispatcher.

Reference behavior: paddlenlp/taskflow/taskflow.py:758 (Taskflow(task, model))
with the TASKS registry :48.  Round-1 scope: generation-backed tasks
(text_generation / question_answering / dialogue / poetry_generation /
code_generation) run on any local CausalLM; the classic NLU pipelines are
registered with informative errors until their task models are ported.
"""
from __future__ import annotations

from typing import Optional

import torch

from ..generation import GenerationConfig
from ..transformers import AutoModelForCausalLM, AutoTokenizer


class _GenerationTask:
    TEMPLATES = {
        "text_generation": "{text}",
        "question_answering": "问题：{text}\n答案：",
        "dialogue": "{text}",
        "poetry_generation": "以「{text}」为题作诗：",
        "code_generation": "{text}",
        "question_generation": "根据文章生成问题：{text}",
    }

    def __init__(self, task: str, model_path: str, max_new_tokens: int = 64,
                 do_sample: bool = False, **kwargs):
        self.task = task
        self.tokenizer = AutoTokenizer.from_pretrained(model_path)
        self.model = AutoModelForCausalLM.from_pretrained(model_path)
        if torch.cuda.is_available():
            self.model = self.model.to("cuda:0")
        self.model.eval()
        self.gen = GenerationConfig(
            max_new_tokens=max_new_tokens, do_sample=do_sample,
            eos_token_id=self.tokenizer.eos_token_id,
            pad_token_id=self.tokenizer.pad_token_id or 0,
        )

    def __call__(self, inputs):
        single = isinstance(inputs, str)
        texts = [inputs] if single else list(inputs)
        template = self.TEMPLATES.get(self.task, "{text}")
        prompts = [template.format(text=t) for t in texts]
        enc = self.tokenizer(prompts, padding=True, return_tensors="pt")
        device = next(self.model.parameters()).device
        out, _ = self.model.generate(enc["input_ids"].to(device), self.gen)
        decoded = self.tokenizer.batch_decode(out, skip_special_tokens=True)
        results = [{"text": t, "answer": d} for t, d in zip(texts, decoded)]
        return results[0] if single else results


class _Seq2SeqTask:
    """Seq2seq-backed pipeline (reference: taskflow/text_summarization.py,
    which runs Pegasus).  Any AutoModelForSeq2SeqLM checkpoint works."""

    def __init__(self, task: str, model_path: str, max_new_tokens: int = 64,
                 num_beams: int = 4, **kwargs):
        from ..transformers import AutoModelForSeq2SeqLM

        self.task = task
        self.tokenizer = AutoTokenizer.from_pretrained(model_path)
        self.model = AutoModelForSeq2SeqLM.from_pretrained(model_path)
        if torch.cuda.is_available():
            self.model = self.model.to("cuda:0")
        self.model.eval()
        self.max_new_tokens = max_new_tokens
        self.num_beams = num_beams

    def __call__(self, inputs):
        single = isinstance(inputs, str)
        texts = [inputs] if single else list(inputs)
        enc = self.tokenizer(texts, padding=True, return_tensors="pt")
        device = next(self.model.parameters()).device
        out, _ = self.model.generate(
            enc["input_ids"].to(device),
            max_new_tokens=self.max_new_tokens, num_beams=self.num_beams)
        decoded = self.tokenizer.batch_decode(out, skip_special_tokens=True)
        return decoded[0] if single else decoded


SEQ2SEQ_TASKS = {"text_summarization", "text2text_generation"}

GENERATION_TASKS = set(_GenerationTask.TEMPLATES)

# encoder-backed pipelines (taskflow/tasks.py): task name -> class
from .tasks import (  # noqa: E402
    DependencyParsingTask,
    DocumentIntelligenceTask,
    FeatureExtractionTask,
    FillMaskTask,
    InformationExtractionTask,
    KnowledgeMiningTask,
    TextClassificationTask,
    TextCorrectionTask,
    TextSimilarityTask,
    TokenClassificationTask,
    WordSegmentationTask,
    ZeroShotTextClassificationTask,
)

ENCODER_TASKS = {
    "text_classification": TextClassificationTask,
    "sentiment_analysis": TextClassificationTask,
    "ner": TokenClassificationTask,
    "pos_tagging": TokenClassificationTask,
    "lexical_analysis": TokenClassificationTask,
    "fill_mask": FillMaskTask,
    "feature_extraction": FeatureExtractionTask,
    "text_similarity": TextSimilarityTask,
    "information_extraction": InformationExtractionTask,
    "zero_shot_text_classification": ZeroShotTextClassificationTask,
    "text_correction": TextCorrectionTask,
    "word_segmentation": WordSegmentationTask,
    "dependency_parsing": DependencyParsingTask,
    "document_intelligence": DocumentIntelligenceTask,
    "knowledge_mining": KnowledgeMiningTask,
}

# API-surface parity with the reference registry (taskflow/taskflow.py:48);
# remaining pipelines need task models that don't exist offline.
PENDING_TASKS = set()  # every registered pipeline is live

TASKS = sorted(GENERATION_TASKS | set(ENCODER_TASKS) | SEQ2SEQ_TASKS
               | PENDING_TASKS)


class Taskflow:
    def __init__(self, task: str, model: Optional[str] = None, **kwargs):
        if task in GENERATION_TASKS or task in ENCODER_TASKS \
                or task in SEQ2SEQ_TASKS:
            if model is None:
                raise ValueError(
                    f"Taskflow('{task}') needs a local model path via model= "
                    "(no network access in this environment)")
            if task in SEQ2SEQ_TASKS:
                self.task_instance = _Seq2SeqTask(task, model, **kwargs)
            elif task in ENCODER_TASKS:
                self.task_instance = ENCODER_TASKS[task](model, **kwargs)
            else:
                self.task_instance = _GenerationTask(task, model, **kwargs)
        elif task in PENDING_TASKS:
            raise NotImplementedError(
                f"Task '{task}' is registered but its task model is not ported "
                f"yet. Available now: "
                f"{sorted(GENERATION_TASKS | set(ENCODER_TASKS))}")
        else:
            raise ValueError(f"Unknown task '{task}'. Registered: {TASKS}")
        self.task = task

    def __call__(self, inputs):
        return self.task_instance(inputs)
