#!/usr/bin/env python3
"""Pre-download model weights onto a PV (reference
scripts/huggingface_downloader.py counterpart): fetches a repo's
safetensors + tokenizer into the layout
production_stack_amd.engine.weights.load_safetensors reads, so engine
pods with modelSpec.pvcStorage start without network access."""

import argparse
import sys


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("model_id", help="e.g. meta-llama/Llama-3.1-8B")
    ap.add_argument("output_dir")
    ap.add_argument("--revision", default=None)
    ap.add_argument("--token", default=None,
                    help="HF token (or set HF_TOKEN)")
    args = ap.parse_args()
    try:
        from huggingface_hub import snapshot_download
    except ImportError:
        print("huggingface_hub is required", file=sys.stderr)
        return 1
    path = snapshot_download(
        args.model_id,
        revision=args.revision,
        token=args.token,
        local_dir=args.output_dir,
        allow_patterns=["*.safetensors", "*.json", "tokenizer*",
                        "*.model"],
    )
    print(f"downloaded to {path}")
    return 0


# ---------------------------------------------------------------------------
# sidecar mode (reference docker/Dockerfile.sidecar): a small HTTP
# service the LoRA controller / init containers call to fetch weights
# or adapters onto a shared volume.
try:
    from fastapi import FastAPI

    app = FastAPI(title="weights downloader sidecar")

    @app.get("/health")
    async def health():
        return {"status": "ok"}

    @app.post("/download")
    async def download(body: dict):
        from huggingface_hub import snapshot_download

        path = snapshot_download(
            body["model_id"],
            revision=body.get("revision"),
            token=body.get("token"),
            local_dir=body.get("target_dir"),
            allow_patterns=body.get(
                "allow_patterns",
                ["*.safetensors", "*.json", "tokenizer*", "*.model"]),
        )
        return {"path": path}
except ImportError:  # CLI-only environments
    app = None


if __name__ == "__main__":
    sys.exit(main())
