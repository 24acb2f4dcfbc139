"""Typer CLI: the user-facing verbs of the framework.

Reference parity: distllm/cli.py (embed, merge, generate, tokenize,
chunk-fasta-file) with the same flag names; plus chat / serve / eval
entry points for the L6 applications.
Entry point: `distllm = distllm_amd.cli:main` (pyproject) or
`python -m distllm_amd.cli`.
"""

from __future__ import annotations

from pathlib import Path

import typer

app = typer.Typer(no_args_is_help=True, pretty_exceptions_enable=False)


@app.callback(invoke_without_command=True)
def _main_options(
    version: bool = typer.Option(
        False, "--version", "-V", is_eager=True,
        help="print the distllm-amd version and exit"),
) -> None:
    if version:
        import distllm_amd

        typer.echo(f"distllm-amd {distllm_amd.__version__}")
        raise typer.Exit()


@app.command()
def embed(
    # both the reference's snake_case spellings + short aliases
    # (reference cli.py:15-128) and the dash-case typer default work
    encoder_name: str = typer.Option(
        "auto", "--encoder-name", "--encoder_name", "-mn",
        help="auto|esm2|esmc"),
    pretrained_model_name_or_path: str = typer.Option(
        "pubmedbert-base", "--pretrained-model-name-or-path",
        "--pretrained_model_name_or_path", "-m",
        help="model preset or local checkpoint"),
    data_path: Path = typer.Option(
        ..., "--data-path", "--data_path", "-d",
        help="input file or directory"),
    data_extension: str = typer.Option(
        "jsonl", "--data-extension", "--data_extension", "-de"),
    output_path: Path = typer.Option(
        ..., "--output-path", "--output_path", "-o"),
    dataset_name: str = typer.Option(
        "jsonl_chunk", "--dataset-name", "--dataset_name", "-dn",
        help="jsonl|jsonl_chunk|fasta|sequence_per_line|huggingface"),
    batch_size: int = typer.Option(
        8, "--batch-size", "--batch_size", "-b"),
    chunk_batch_size: int = typer.Option(
        8, "--chunk-batch-size", "--chunk_batch_size", "-cb"),
    buffer_size: int = typer.Option(
        1, "--buffer-size", "--buffer_size", "-bs"),
    pooler_name: str = typer.Option(
        "mean", "--pooler-name", "--pooler_name", "-pn",
        help="mean|last_token"),
    embedder_name: str = typer.Option(
        "semantic_chunk", "--embedder-name", "--embedder_name", "-en",
        help="full_sequence|semantic_chunk"),
    writer_name: str = typer.Option(
        "huggingface", "--writer-name", "--writer_name", "-wn",
        help="huggingface|numpy"),
    half_precision: bool = typer.Option(
        True, "--half-precision/--no-half-precision",
        "--half_precision/--no-half_precision", "-hp"),
    eval_mode: bool = typer.Option(
        True, "--eval-mode/--no-eval-mode",
        "--eval_mode/--no-eval_mode", "-em"),
    compile_model: bool = typer.Option(
        False, "--compile-model/--no-compile-model",
        "--compile_model/--no-compile_model", "-cm"),
    quantization: bool = typer.Option(
        False, "--quantization/--no-quantization", "-q"),
) -> None:
    """Embed a file/directory serially in-process (single GPU)."""
    from .distributed_embedding import embedding_worker

    dataset_kwargs: dict = {"name": dataset_name, "batch_size": batch_size}
    if dataset_name == "jsonl_chunk":
        dataset_kwargs["buffer_size"] = buffer_size
    encoder_kwargs = {
        "name": encoder_name,
        "pretrained_model_name_or_path": pretrained_model_name_or_path,
        "half_precision": half_precision,
    }
    embedder_kwargs: dict = {"name": embedder_name}
    if embedder_name == "semantic_chunk":
        embedder_kwargs["chunk_batch_size"] = chunk_batch_size

    files = (
        [data_path]
        if data_path.is_file()
        else sorted(data_path.glob(f"*.{data_extension}"))
    )
    output_path.mkdir(exist_ok=True, parents=True)
    for f in files:
        embedding_worker(
            f, output_path, dataset_kwargs, encoder_kwargs,
            {"name": pooler_name}, embedder_kwargs, {"name": writer_name},
        )


@app.command()
def merge(
    writer_name: str = typer.Option(
        "huggingface", "--writer-name", "--writer_name", "-wn"),
    num_proc: int = typer.Option(
        1, "--num-proc", "--num_proc", "-np"),
    dataset_dir: Path = typer.Option(
        ..., "--dataset-dir", "--dataset_dir", "-d",
        help="dir of dataset shards"),
    output_dir: Path = typer.Option(
        ..., "--output-dir", "--output_dir", "-o"),
) -> None:
    """Merge per-file dataset shards into one dataset."""
    from .embed import get_writer

    writer = get_writer({"name": writer_name})
    shards = sorted(p for p in Path(dataset_dir).iterdir() if p.is_dir())
    writer.merge(shards, output_dir, num_proc=num_proc)
    typer.echo(f"merged {len(shards)} shards -> {output_dir}")


@app.command()
def generate(
    input_dir: Path = typer.Option(
        ..., "--input-dir", "--input_dir", "-i"),
    output_dir: Path = typer.Option(
        ..., "--output-dir", "--output_dir", "-o"),
    prompt_name: str = typer.Option(
        "question_chunk", "--prompt-name", "--prompt_name", "-pn"),
    reader_name: str = typer.Option(
        "huggingface", "--reader-name", "--reader_name", "-rn"),
    writer_name: str = typer.Option(
        "huggingface", "--writer-name", "--writer_name", "-wn"),
    generator_name: str = typer.Option(
        "native", "--generator-name", "--generator_name", "-gn",
        help="native|vllm|openai_http"),
    llm_name: str = typer.Option(
        "llama3-8b", "--llm-name", "--llm_name", "-vmn"),
    temperature: float = typer.Option(
        0.5, "--temperature", "-t"),
    min_p: float = typer.Option(0.1, "--min-p", "--min_p", "-mp"),
    top_p: float = typer.Option(0.0, "--top-p", "--top_p", "-tp"),
    max_tokens: int = typer.Option(
        2000, "--max-tokens", "--max_tokens", "-mt"),
    use_beam_search: bool = typer.Option(
        False, "--use-beam-search/--no-use-beam-search",
        "--use_beam_search/--no-use_beam_search",
        help="beam search (reference vLLM flag; num_beams=10)"),
    batch_size: int = typer.Option(
        64, "--batch-size", "--batch_size", "-b"),
    quantization: bool = typer.Option(
        False, "--quantization/--no-quantization", "-q",
        help="W8A16 fp8-e4m3 decode weights on GPU"),
) -> None:
    """Generate over a directory of inputs serially in-process."""
    from .distributed_generation import generate_worker

    generator_kwargs = {
        "name": generator_name,
        "model": llm_name,
        "temperature": temperature,
        "min_p": min_p,
        "top_p": top_p,
        "max_tokens": max_tokens,
        "max_batch_size": batch_size,
        "quantization": quantization,
    }
    if use_beam_search:
        generator_kwargs["num_beams"] = 10
    if generator_name in ("openai_http", "argo"):
        generator_kwargs = {"name": generator_name, "model": llm_name,
                            "temperature": temperature,
                            "max_tokens": max_tokens}
    elif generator_name in ("langchain", "api"):
        generator_kwargs = {"name": generator_name, "llm": llm_name,
                            "temperature": temperature,
                            "max_tokens": max_tokens}
    output_dir.mkdir(parents=True, exist_ok=True)
    input_dir = Path(input_dir)
    if (input_dir / "state.json").exists() or (
            input_dir / "dataset_info.json").exists():
        # input_dir IS one saved HF dataset (e.g. `distllm merge` output),
        # not a directory of shards: process it as a single input.
        inputs = [input_dir]
    else:
        inputs = sorted(input_dir.iterdir())
    for f in inputs:
        generate_worker(f, output_dir, {"name": prompt_name},
                        {"name": reader_name}, {"name": writer_name},
                        generator_kwargs)


@app.command()
def tokenize(
    input_dir: Path = typer.Option(
        ..., "--input-dir", "--input_dir", "-i"),
    output_dir: Path = typer.Option(
        ..., "--output-dir", "--output_dir", "-o"),
    text_field: str = typer.Option(
        "text", "--text-field", "--text_field", "-tf"),
    tokenizer_name: str = typer.Option(
        "pubmedbert-base", "--tokenizer-name", "--tokenizer_name",
        "-tn"),
    dotenv_path: Path = typer.Option(
        None, "--dotenv-path", "--dotenv_path", "-dp",
        help="env file loaded before tokenizing (reference flag)"),
    save_labels: bool = typer.Option(
        False, "--save-labels/--no-save-labels",
        "--save_labels/--no-save_labels", "-sl"),
) -> None:
    """Tokenize jsonl files into HF datasets."""
    from .distributed_tokenization import TokenizationConfig, main as tmain

    if dotenv_path is not None:
        from .generate.generators import ApiGenerator

        ApiGenerator._load_dotenv(str(dotenv_path))
    tmain(TokenizationConfig(
        input_dir=input_dir, output_dir=output_dir, text_field=text_field,
        tokenizer_name=tokenizer_name, save_labels=save_labels,
    ))


@app.command()
def chunk_fasta_file(
    input_file: Path = typer.Option(
        ..., "--input-file", "--input_file", "-i"),
    output_dir: Path = typer.Option(
        ..., "--output-dir", "--output_dir", "-o"),
    num_chunks: int = typer.Option(
        8, "--num-chunks", "--chunk_size", "-c",
        help="number of smaller files (reference --chunk_size)"),
) -> None:
    """Split one fasta file into N chunk files."""
    from .embed.datasets import read_fasta, write_fasta
    from .utils import batch_data

    seqs = read_fasta(input_file)
    output_dir.mkdir(parents=True, exist_ok=True)
    per = max(1, (len(seqs) + num_chunks - 1) // num_chunks)
    for i, chunk in enumerate(batch_data(seqs, per)):
        write_fasta(chunk, output_dir / f"{input_file.stem}_{i}.fasta")
    typer.echo(f"wrote {min(num_chunks, len(seqs))} chunks")


@app.command()
def chat(
    config: Path = typer.Option(..., help="chat YAML config"),
) -> None:
    """Terminal RAG chat."""
    from .chat import chat_with_model, ChatAppConfig

    chat_with_model(ChatAppConfig.from_yaml(config))


@app.command()
def serve(
    config: Path = typer.Option(..., help="chat YAML config"),
    host: str = typer.Option("0.0.0.0"),
    port: int = typer.Option(8000),
) -> None:
    """OpenAI-compatible RAG chat server."""
    import os

    import uvicorn

    os.environ["DISTLLM_CHAT_CONFIG"] = str(config)
    from .chat_server import app as fastapi_app

    uvicorn.run(fastapi_app, host=host, port=port)


@app.command()
def evaluate(
    config: Path = typer.Option(..., help="eval-suite YAML config"),
) -> None:
    """Run the RAG eval suite."""
    from .rag.evaluate import EvalSuiteConfig, run_eval_suite

    run_eval_suite(EvalSuiteConfig.from_yaml(config))


@app.command()
def launch(
    config: Path = typer.Option(..., help="HpcLaunchConfig YAML"),
    command: str = typer.Option(
        ..., help="module command to run under torchrun, e.g. "
        "'-m distllm_amd.distributed_embedding --config embed.yaml'"),
    output: Path = typer.Option(
        None, help="write the batch script here (default: stdout)"),
) -> None:
    """Render a Slurm/PBS batch script for a multi-node torchrun job
    (the reference's Parsl HPC platform role). Accepts either an
    HpcLaunchConfig YAML or a full pipeline YAML whose compute_config
    names a reference HPC platform (polaris/leonardo/slurm/pbs)."""
    import yaml as _yaml

    from .parallel.launcher import HpcLaunchConfig, render_batch_script
    from .parallel.runtime import ComputeConfig

    raw = _yaml.safe_load(Path(config).read_text())
    if isinstance(raw, dict) and "compute_config" in raw:
        lc = ComputeConfig(**raw["compute_config"]).to_launch_config()
        if lc is None:
            raise typer.BadParameter(
                "compute_config names an in-process platform; use "
                "name: polaris/leonardo/slurm/pbs for batch launch")
    else:
        lc = HpcLaunchConfig(**raw)
    script = render_batch_script(lc, command)
    if output is not None:
        output.write_text(script)
        typer.echo(f"wrote {output}")
    else:
        typer.echo(script)


def main() -> None:
    app()


if __name__ == "__main__":
    main()
