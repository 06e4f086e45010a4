#!/usr/bin/env python3
"""Model launcher — registry + download + run-script generation.

Role parity with the reference launch.py (registry of 11 prebuilt Q40
models on HF, multi-part downloads concatenated into one .m, run scripts).
Multi-part models are one .m split into chunks suffixed _aa, _ab, ... that
concatenate in order. This environment has no network, so when download
fails the offline converter path is printed instead.

Usage:
  python launch.py                      # list models
  python launch.py qwen3_8b_q40         # download + write run_*.sh
"""

import os
import sys
import time
import urllib.request

HF = "https://huggingface.co/b4rtaz"


def _parts(n: int) -> list[str]:
    """Chunk suffixes aa, ab, ac, ... (reference multi-part naming)."""
    return [chr(97 + i // 26) + chr(97 + i % 26) for i in range(n)]


def _multi(repo: str, stem: str, n: int) -> list[str]:
    return [f"{HF}/{repo}/resolve/main/{stem}_{s}?download=true"
            for s in _parts(n)]


def _single(repo: str, fname: str) -> list[str]:
    return [f"{HF}/{repo}/resolve/main/{fname}?download=true"]


MODELS = {
    "llama3_1_8b_instruct_q40": {
        "model": _single("Llama-3_1-8B-Q40-Instruct-Distributed-Llama",
                         "dllama_model_llama3.1_instruct_q40.m"),
        "tokenizer": _single("Llama-3_1-8B-Q40-Instruct-Distributed-Llama",
                             "dllama_tokenizer_llama3_1.t"),
        "size": "6.32 GB", "arch": "llama-3.1-8b",
    },
    "llama3_1_405b_instruct_q40": {
        "model": _multi("Llama-3_1-405B-Q40-Instruct-Distributed-Llama",
                        "dllama_model_llama31_405b_q40", 56),
        "tokenizer": _single("Llama-3_1-405B-Q40-Instruct-Distributed-Llama",
                             "dllama_tokenizer_llama_3_1.t"),
        "size": "238 GB", "arch": "llama-3.1-405b",
    },
    "llama3_2_1b_instruct_q40": {
        "model": _single("Llama-3_2-1B-Instruct-Q40-Distributed-Llama",
                         "dllama_model_llama3.2-1b-instruct_q40.m"),
        "tokenizer": _single("Llama-3_2-1B-Instruct-Q40-Distributed-Llama",
                             "dllama_tokenizer_llama3_2-1b-instruct.t"),
        "size": "1.7 GB", "arch": "llama-3.2-1b",
    },
    "llama3_2_3b_instruct_q40": {
        "model": _single("Llama-3_2-3B-Instruct-Q40-Distributed-Llama",
                         "dllama_model_llama3.2-3b-instruct_q40.m"),
        "tokenizer": _single("Llama-3_2-3B-Instruct-Q40-Distributed-Llama",
                             "dllama_tokenizer_llama3_2-3b-instruct.t"),
        "size": "3.4 GB", "arch": "llama-3.2-3b",
    },
    "llama3_3_70b_instruct_q40": {
        "model": _single("Llama-3_3-70B-Instruct-Q40-Distributed-Llama",
                         "dllama_model_llama-3.3-70b_q40.m"),
        "tokenizer": _single("Llama-3_3-70B-Instruct-Q40-Distributed-Llama",
                             "dllama_tokenizer_llama-3.3-70b.t"),
        "size": "40 GB", "arch": "llama-3.3-70b",
    },
    "deepseek_r1_distill_llama_8b_q40": {
        "model": _single("DeepSeek-R1-Distill-Llama-8B-Distributed-Llama",
                         "dllama_model_deepseek-r1-distill-llama-8b_q40.m"),
        "tokenizer": _single("DeepSeek-R1-Distill-Llama-8B-Distributed-Llama",
                             "dllama_tokenizer_deepseek-r1-distill-llama-8b.t"),
        "size": "6.32 GB", "arch": "llama-3.1-8b",
    },
    "qwen3_0.6b_q40": {
        "model": _single("Qwen3-0.6B-Q40-Distributed-Llama",
                         "dllama_model_qwen3_0.6b_q40.m"),
        "tokenizer": _single("Qwen3-0.6B-Q40-Distributed-Llama",
                             "dllama_tokenizer_qwen3_0.6b.t"),
        "size": "0.9 GB", "arch": "qwen3-0.6b",
    },
    "qwen3_1.7b_q40": {
        "model": _single("Qwen3-1.7B-Q40-Distributed-Llama",
                         "dllama_model_qwen3_1.7b_q40.m"),
        "tokenizer": _single("Qwen3-1.7B-Q40-Distributed-Llama",
                             "dllama_tokenizer_qwen3_1.7b.t"),
        "size": "1.6 GB", "arch": "qwen3-1.7b",
    },
    "qwen3_8b_q40": {
        "model": _single("Qwen3-8B-Q40-Distributed-Llama",
                         "dllama_model_qwen3_8b_q40.m"),
        "tokenizer": _single("Qwen3-8B-Q40-Distributed-Llama",
                             "dllama_tokenizer_qwen3_8b.t"),
        "size": "5.9 GB", "arch": "qwen3-8b",
    },
    "qwen3_14b_q40": {
        "model": _multi("Qwen3-14B-Q40-Distributed-Llama",
                        "dllama_model_qwen3_14b_q40", 2),
        "tokenizer": _single("Qwen3-14B-Q40-Distributed-Llama",
                             "dllama_tokenizer_qwen3_14b.t"),
        "size": "10.2 GB", "arch": "qwen3-14b",
    },
    "qwen3_30b_a3b_q40": {
        "model": _multi("Qwen3-30B-A3B-Q40-Distributed-Llama",
                        "dllama_model_qwen3_30b_a3b", 5),
        "tokenizer": _single("Qwen3-30B-A3B-Q40-Distributed-Llama",
                             "dllama_tokenizer_qwen3_30b_a3b.t"),
        "size": "17 GB", "arch": "qwen3-30b-a3b",
    },
}


def download_concat(urls: list[str], path: str, retries: int = 8) -> bool:
    """Download URLs in order, concatenated into one file at `path`.
    A failed part is retried from its own start offset (seek+truncate),
    so earlier completed parts are never refetched."""
    if os.path.exists(path):
        print(f"  ✅ {path} (exists)")
        return True
    tmp = path + ".part"
    try:
        with open(tmp, "wb") as f:
            for url in urls:
                start = f.tell()
                for attempt in range(retries):
                    print(f"  📥 {url}" + (f" (retry {attempt})" if attempt else ""))
                    try:
                        with urllib.request.urlopen(url, timeout=30) as r:
                            while True:
                                chunk = r.read(1 << 20)
                                if not chunk:
                                    break
                                f.write(chunk)
                        break
                    except Exception as e:  # noqa: BLE001
                        print(f"  ⚠️  {e}")
                        f.seek(start)
                        f.truncate()
                        time.sleep(attempt)
                else:
                    raise OSError(f"failed after {retries} attempts: {url}")
        os.rename(tmp, path)
        return True
    except Exception as e:  # noqa: BLE001
        print(f"  ⚠️  download failed ({e}); no network? Convert offline with:\n"
              f"     python converter/convert_hf.py <hf_checkpoint_dir> q40 <name>\n"
              f"     python converter/convert_tokenizer_hf.py <hf_checkpoint_dir> <name>")
        if os.path.exists(tmp):
            os.remove(tmp)
        return False


def write_run_script(name: str, model_path: str, tok_path: str) -> str:
    script = f"run_{name}.sh"
    with open(script, "w") as f:
        f.write(f"""#!/bin/sh
# single GPU
python -m dllama_amd.apps.main inference --model {model_path} \\
    --tokenizer {tok_path} --prompt "Hello world" --steps 64
# all 8 GPUs of one MI355X node (TP=8 over RCCL/xGMI):
# torchrun --nproc-per-node 8 --master-addr 127.0.0.1 -m dllama_amd.apps.main \\
#     inference --model {model_path} --tokenizer {tok_path} --prompt "Hello" --steps 64
""")
    os.chmod(script, 0o755)
    return script


def main():
    if len(sys.argv) < 2:
        print("Usage: python launch.py <model>\n\nAvailable models:")
        for k, v in MODELS.items():
            np = len(v["model"])
            print(f"  {k:34s} {v['size']:>8s}"
                  + (f"  ({np} parts)" if np > 1 else ""))
        return 0
    name = sys.argv[1]
    if name not in MODELS:
        print(f"unknown model {name}")
        return 1
    spec = MODELS[name]
    model_path = f"models/{name}/dllama_model_{name}.m"
    tok_path = f"models/{name}/dllama_tokenizer_{name}.t"
    os.makedirs(os.path.dirname(model_path), exist_ok=True)
    ok = download_concat(spec["model"], model_path) and \
        download_concat(spec["tokenizer"], tok_path)
    script = write_run_script(name, model_path, tok_path)
    print(f"📄 wrote {script}" + ("" if ok else " (files still missing)"))
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
