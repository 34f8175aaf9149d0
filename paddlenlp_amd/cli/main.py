"""`paddlenlp_amd` CLI (reference: paddlenlp/cli/main.py:99-243 typer app).

Subcommands: search (list registered models), server (serve a model),
convert (HF safetensors dir -> framework layout check), download (stub —
no network in this environment).
"""
from __future__ import annotations

import sys

try:
    import typer
except ImportError:  # pragma: no cover
    typer = None

from ..transformers.auto.registry import MODEL_REGISTRY


def _build_app():
    app = typer.Typer(help="paddlenlp_amd command line")

    @app.command()
    def search(query: str = typer.Argument("", help="filter model types")):
        """List registered model families."""
        for mt, entry in sorted(MODEL_REGISTRY.items()):
            if query and query not in mt:
                continue
            main_cls = entry.get("causal_lm") or entry.get("seq2seq_lm") \
                or entry.get("base")
            typer.echo(f"{mt:<14} {main_cls} ({entry['module']})")

    @app.command()
    def server(model: str = typer.Option(..., help="local model dir"),
               port: int = typer.Option(8189),
               max_new_tokens: int = typer.Option(64)):
        """Serve a local CausalLM over HTTP."""
        from ..server import SimpleServer
        from ..taskflow import Taskflow

        flow = Taskflow("text_generation", model=model, max_new_tokens=max_new_tokens)
        srv = SimpleServer()
        srv.register_taskflow("/taskflow/text_generation", flow)
        srv.run(port=port)

    @app.command()
    def convert(path: str = typer.Argument(..., help="model dir to validate")):
        """Validate a local model directory loads in this framework."""
        from ..transformers import AutoConfig, AutoModelForCausalLM

        cfg = AutoConfig.from_pretrained(path)
        typer.echo(f"model_type={cfg.model_type}")
        model = AutoModelForCausalLM.from_pretrained(path)
        n = sum(p.numel() for p in model.parameters())
        typer.echo(f"loaded OK: {n/1e6:.1f}M params")

    @app.command()
    def download(model: str = typer.Argument(...)):
        """(no network in this environment)"""
        typer.echo("This environment has no network access; place model files "
                   "in a local directory and pass its path instead.")
        raise typer.Exit(1)

    return app


def main():
    if typer is None:
        print("typer is not installed", file=sys.stderr)
        sys.exit(1)
    _build_app()()


if __name__ == "__main__":
    main()
