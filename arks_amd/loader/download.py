"""Model-cache download worker (the job the ArksModel controller's one-shot
pod runs; contract from reference scripts/download.py +
arksmodel_controller.go:218-359):

  env MODEL_NAME  - HuggingFace repo id (e.g. Qwen/Qwen2.5-7B-Instruct)
      MODEL_PATH  - target dir on the shared PVC (/models/models/<ns>/<name>)
      HF_TOKEN    - optional hub token

Exit code drives the pod phase (0 -> Loaded, else Failed with the last error
line as the termination message). Transient failures retry with backoff;
downloads resume (hub cache) so a retried pod does not restart from zero.

Run: python3 -m arks_amd.loader.download
"""

from __future__ import annotations

import os
import sys
import time

RETRIES = 3
RETRY_DELAY_S = 10


def fail(msg: str) -> "NoReturn":  # noqa: F821
    print(f"download error: {msg}", file=sys.stderr, flush=True)
    sys.exit(1)


def main() -> None:
    model = os.environ.get("MODEL_NAME")
    path = os.environ.get("MODEL_PATH")
    token = os.environ.get("HF_TOKEN") or None
    if not model:
        fail("MODEL_NAME is required")
    if not path:
        fail("MODEL_PATH is required")

    try:
        from huggingface_hub import HfApi, snapshot_download
    except ImportError:
        fail("huggingface_hub is not installed in the downloader image")

    api = HfApi(token=token)
    if token:
        try:
            who = api.whoami()
            print(f"authenticated as {who.get('name', '?')}", flush=True)
        except Exception as e:
            fail(f"HF_TOKEN is invalid: {e}")
    try:
        if not api.repo_exists(model):
            fail(f"model repo {model!r} does not exist "
                 "(or requires a token with access)")
    except SystemExit:
        raise
    except Exception as e:
        print(f"repo existence check failed ({e}); proceeding", flush=True)

    os.makedirs(path, exist_ok=True)
    last_err: Exception | None = None
    for attempt in range(1, RETRIES + 1):
        try:
            print(f"downloading {model} -> {path} (attempt {attempt}/{RETRIES})",
                  flush=True)
            snapshot_download(
                repo_id=model,
                local_dir=path,
                token=token,
                max_workers=8,
            )
            print("download complete", flush=True)
            return
        except Exception as e:  # noqa: BLE001 — retry any hub/network error
            last_err = e
            print(f"attempt {attempt} failed: {e}", file=sys.stderr, flush=True)
            if attempt < RETRIES:
                time.sleep(RETRY_DELAY_S)
    fail(f"download failed after {RETRIES} attempts: {last_err}")


if __name__ == "__main__":
    main()
