"""Distributed generation entry point.

Reference parity: distllm/distributed_generation.py (same YAML schema:
input_dir, output_dir, glob_patterns, prompt_config, reader_config,
writer_config, generator_config, compute_config; the output dir must
not pre-exist), with Parsl replaced by the per-rank RCCL shard.
"""

from __future__ import annotations

import argparse
from pathlib import Path
from typing import Any

from .generate import (
    get_generate_writer,
    get_generator,
    get_prompt_template,
    get_reader,
)
from .parallel import barrier, init_distributed, shard_list
from .utils import BaseConfig, Timer


class GenerationConfig(BaseConfig):
    input_dir: Path
    output_dir: Path
    glob_patterns: list[str] = ["*"]
    prompt_config: dict[str, Any]
    reader_config: dict[str, Any]
    writer_config: dict[str, Any]
    generator_config: dict[str, Any]
    compute_config: dict[str, Any] = {"name": "local"}


def generate_worker(
    input_path: Path,
    output_dir: Path,
    prompt_kwargs: dict[str, Any],
    reader_kwargs: dict[str, Any],
    writer_kwargs: dict[str, Any],
    generator_kwargs: dict[str, Any],
) -> None:
    """Generate over one input file (reference
    distributed_generation.py:22-86 structure)."""
    with Timer("loaded-generator", str(input_path)):
        generator = get_generator(generator_kwargs, register=True)
    reader = get_reader(reader_kwargs)
    prompt = get_prompt_template(prompt_kwargs)
    writer = get_generate_writer(writer_kwargs)

    with Timer("read-input", str(input_path)):
        text, paths = reader.read(input_path)

    with Timer("generated", str(input_path)):
        prompts = prompt.preprocess(text)
        responses = generator.generate(prompts)
        results = prompt.postprocess(responses)

    # drop empty responses (reference distributed_generation.py:69-75)
    keep = [i for i, r in enumerate(results) if r]
    paths = [paths[i] for i in keep]
    text = [text[i] for i in keep]
    results = [results[i] for i in keep]

    if not results:
        # every response was dropped by postprocess (e.g. none parsed):
        # write nothing — a 0-row HF dataset dir is not even re-loadable,
        # and the merge step would have to skip it anyway.
        print(f"[generate] {input_path}: 0 usable responses, no shard")
        return

    with Timer("wrote-output", str(input_path)):
        from .embed.writers import unique_shard_dir

        writer.write(unique_shard_dir(output_dir), paths, text, results)


def main(config: GenerationConfig) -> None:
    rank, world = init_distributed()
    if rank == 0:
        if config.output_dir.exists():
            raise ValueError(
                f"output dir {config.output_dir} already exists "
                "(reference semantics: refuse to overwrite)"
            )
        config.output_dir.mkdir(parents=True)
        config.write_yaml(config.output_dir / "config.yaml")
    barrier()

    files: list[Path] = []
    for pattern in config.glob_patterns:
        files.extend(sorted(config.input_dir.glob(pattern)))
    my_files = shard_list(files)
    print(f"[rank {rank}/{world}] generating over {len(my_files)} of "
          f"{len(files)} files")
    out = config.output_dir / "generations"
    out.mkdir(exist_ok=True, parents=True)
    for path in my_files:
        with Timer("finished-generation", str(path), f"rank{rank}"):
            generate_worker(path, out, config.prompt_config,
                            config.reader_config, config.writer_config,
                            config.generator_config)
    barrier()


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("--config", type=Path, required=True)
    args = parser.parse_args()
    main(GenerationConfig.from_yaml(args.config))
