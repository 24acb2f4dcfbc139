"""Writer strategies for the generate pipeline.

Reference parity: distllm/generate/writers/* (huggingface, amp_json):
write(dir, paths, text, responses) + fault-tolerant merge.
"""

from __future__ import annotations

import json
from pathlib import Path
from typing import Any, Literal, Union

from ..utils import BaseConfig


class HuggingFaceGenWriterConfig(BaseConfig):
    name: Literal["huggingface"] = "huggingface"


class HuggingFaceGenWriter:
    def __init__(self, config: HuggingFaceGenWriterConfig):
        self.config = config

    def write(self, output_dir: Path, paths: list[str], text: list[str],
              responses: list[str]) -> None:
        from datasets import Dataset as HFDataset

        ds = HFDataset.from_dict(
            {"path": paths, "text": text, "response": responses}
        )
        Path(output_dir).mkdir(parents=True, exist_ok=True)
        ds.save_to_disk(str(output_dir))

    def merge(self, dataset_dirs: list[Path], output_dir: Path,
              num_proc: int = 1) -> None:
        from datasets import concatenate_datasets, load_from_disk

        parts = []
        for d in dataset_dirs:
            try:
                parts.append(load_from_disk(str(d)))
            except Exception as e:
                print(f"[merge] skipping {d}: {e}")
        if not parts:
            raise ValueError("no readable dataset shards to merge")
        concatenate_datasets(parts).save_to_disk(
            str(output_dir),
            num_proc=num_proc if num_proc and num_proc > 1 else None,
        )


class AMPJsonlWriterConfig(BaseConfig):
    name: Literal["amp_json"] = "amp_json"
    base_name: str = "question_set"  # reference amp_json.py:21 default


class AMPJsonlWriter:
    """Merge each response back into its original JSON entry (the
    `paths` carry the entry JSON) and append to chunked jsonl files."""

    def __init__(self, config: AMPJsonlWriterConfig):
        self.config = config
        self.current_chunk = 0

    def write(self, output_dir: Path, paths: list[str], text: list[str],
              responses: list[str]) -> None:
        output_dir = Path(output_dir)
        output_dir.mkdir(parents=True, exist_ok=True)
        out = output_dir / f"{self.config.base_name}_{self.current_chunk}.jsonl"
        with open(out, "w") as fh:
            for entry_json, resp in zip(paths, responses):
                try:
                    entry = json.loads(entry_json)
                except json.JSONDecodeError:
                    entry = {"entry": entry_json}
                entry["response"] = resp
                fh.write(json.dumps(entry) + "\n")
        self.current_chunk += 1

    def merge(self, dataset_dirs: list[Path], output_dir: Path,
              num_proc: int = 1) -> None:
        output_dir = Path(output_dir)
        output_dir.mkdir(parents=True, exist_ok=True)
        out = output_dir / f"{self.config.base_name}_merged.jsonl"
        with open(out, "w") as fh:
            for d in dataset_dirs:
                for f in sorted(Path(d).glob("*.jsonl")):
                    fh.write(f.read_text())


GenWriterConfigs = Union[HuggingFaceGenWriterConfig, AMPJsonlWriterConfig]

STRATEGIES: dict[str, tuple[type, type]] = {
    "huggingface": (HuggingFaceGenWriterConfig, HuggingFaceGenWriter),
    "amp_json": (AMPJsonlWriterConfig, AMPJsonlWriter),
}


def get_generate_writer(kwargs: dict[str, Any]):
    name = kwargs.get("name", "")
    if name not in STRATEGIES:
        raise ValueError(
            f"unknown writer name {name!r}; options: {sorted(STRATEGIES)}"
        )
    cfg_cls, impl_cls = STRATEGIES[name]
    return impl_cls(cfg_cls(**kwargs))
