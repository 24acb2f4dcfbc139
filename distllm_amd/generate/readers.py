"""Reader strategies for the generate pipeline.

Reference parity: distllm/generate/readers/* (huggingface, jsonl,
amp_json): read(path) -> (texts, paths).
"""

from __future__ import annotations

import json
from pathlib import Path
from typing import Any, Literal, Union

from ..utils import BaseConfig


class HuggingFaceReaderConfig(BaseConfig):
    name: Literal["huggingface"] = "huggingface"
    text_field: str = "text"
    path_field: str = "path"


class HuggingFaceReader:
    def __init__(self, config: HuggingFaceReaderConfig):
        self.config = config

    def read(self, input_path: Path) -> tuple[list[str], list[str]]:
        from datasets import load_from_disk

        ds = load_from_disk(str(input_path))
        text = list(ds[self.config.text_field])
        if self.config.path_field in ds.column_names:
            paths = [str(p) for p in ds[self.config.path_field]]
        else:
            paths = [str(input_path)] * len(text)
        return text, paths


class JsonlReaderConfig(BaseConfig):
    name: Literal["jsonl"] = "jsonl"
    text_field: str = "text"
    path_field: str = "path"


class JsonlReader:
    def __init__(self, config: JsonlReaderConfig):
        self.config = config

    def read(self, input_path: Path) -> tuple[list[str], list[str]]:
        text, paths = [], []
        for line in Path(input_path).read_text().splitlines():
            if not line.strip():
                continue
            obj = json.loads(line)
            text.append(obj[self.config.text_field])
            paths.append(str(obj.get(self.config.path_field, input_path)))
        return text, paths


class AMPJsonReaderConfig(BaseConfig):
    name: Literal["amp_json"] = "amp_json"


class AMPJsonReader:
    """JSON dict-of-lists; each entry's own JSON doubles as its path so
    downstream writers can merge responses back into the entry."""

    def __init__(self, config: AMPJsonReaderConfig):
        self.config = config

    def read(self, input_path: Path) -> tuple[list[str], list[str]]:
        with open(input_path) as fh:
            data = json.load(fh)
        text, paths = [], []
        for _key, entries in data.items():
            for entry in entries:
                s = json.dumps(entry)
                text.append(s)
                paths.append(s)
        return text, paths


ReaderConfigs = Union[HuggingFaceReaderConfig, JsonlReaderConfig,
                      AMPJsonReaderConfig]

STRATEGIES: dict[str, tuple[type, type]] = {
    "huggingface": (HuggingFaceReaderConfig, HuggingFaceReader),
    "jsonl": (JsonlReaderConfig, JsonlReader),
    "amp_json": (AMPJsonReaderConfig, AMPJsonReader),
}


def get_reader(kwargs: dict[str, Any]):
    name = kwargs.get("name", "")
    if name not in STRATEGIES:
        raise ValueError(
            f"unknown reader name {name!r}; options: {sorted(STRATEGIES)}"
        )
    cfg_cls, impl_cls = STRATEGIES[name]
    return impl_cls(cfg_cls(**kwargs))
