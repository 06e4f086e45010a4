#!/usr/bin/env python3
"""Model launcher — registry + run-script generation.

Role parity with the reference launch.py:17-73 (model registry of prebuilt
Q40 models with multi-part HF URLs, resumable download, run_*.sh scripts).
This environment has no network, so download is attempted only when
requested and the converter path is documented for offline use.

Usage:
  python launch.py                      # list models
  python launch.py llama3_1_8b_instruct_q40   # download (if network) + run script
"""

import os
import sys
import urllib.request

# (model, tokenizer) URL lists; sizes from reference README.md:28-40
HF = "https://huggingface.co/b4rtaz"
MODELS = {
    "llama3_1_8b_instruct_q40": {
        "model": [f"{HF}/Llama-3_1-8B-Q40-Instruct-Distributed-Llama/resolve/main/dllama_model_llama3.1_instruct_q40.m?download=true"],
        "tokenizer": [f"{HF}/Llama-3_1-8B-Q40-Instruct-Distributed-Llama/resolve/main/dllama_tokenizer_llama3_1.t?download=true"],
        "size": "6.32 GB", "arch": "llama-3.1-8b",
    },
    "llama3_2_1b_instruct_q40": {
        "model": [f"{HF}/Llama-3_2-1B-Instruct-Q40-Distributed-Llama/resolve/main/dllama_model_llama3.2-1b-instruct_q40.m?download=true"],
        "tokenizer": [f"{HF}/Llama-3_2-1B-Instruct-Q40-Distributed-Llama/resolve/main/dllama_tokenizer_llama3_2-1b-instruct.t?download=true"],
        "size": "1.7 GB", "arch": "llama-3.2-1b",
    },
    "llama3_2_3b_instruct_q40": {
        "model": [f"{HF}/Llama-3_2-3B-Instruct-Q40-Distributed-Llama/resolve/main/dllama_model_llama3.2-3b-instruct_q40.m?download=true"],
        "tokenizer": [f"{HF}/Llama-3_2-3B-Instruct-Q40-Distributed-Llama/resolve/main/dllama_tokenizer_llama3_2-3b-instruct.t?download=true"],
        "size": "3.4 GB", "arch": "llama-3.2-3b",
    },
    "llama3_3_70b_instruct_q40": {
        "model": [f"{HF}/Llama-3_3-70B-Instruct-Q40-Distributed-Llama/resolve/main/dllama_model_llama-3.3-70b_q40.m?download=true"],
        "tokenizer": [f"{HF}/Llama-3_3-70B-Instruct-Q40-Distributed-Llama/resolve/main/dllama_tokenizer_llama-3.3-70b.t?download=true"],
        "size": "40 GB", "arch": "llama-3.3-70b",
    },
    "qwen3_30b_a3b_q40": {
        "model": [f"{HF}/Qwen3-30B-A3B-Q40-Distributed-Llama/resolve/main/dllama_model_qwen3_30b_a3b_q40.m?download=true"],
        "tokenizer": [f"{HF}/Qwen3-30B-A3B-Q40-Distributed-Llama/resolve/main/dllama_tokenizer_qwen3_30b_a3b.t?download=true"],
        "size": "17 GB", "arch": "qwen3-30b-a3b",
    },
}


def download(url: str, path: str) -> bool:
    if os.path.exists(path):
        print(f"  ✅ {path} (exists)")
        return True
    try:
        print(f"  📥 {url}")
        urllib.request.urlretrieve(url, path + ".part")
        os.rename(path + ".part", path)
        return True
    except Exception as e:  # noqa: BLE001
        print(f"  ⚠️  download failed ({e}); no network? Convert offline with:\n"
              f"     python converter/convert_hf.py <hf_checkpoint_dir> q40 <name>\n"
              f"     python converter/convert_tokenizer_hf.py <hf_checkpoint_dir> <name>")
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
            print(f"  {k:32s} {v['size']}")
        return 0
    name = sys.argv[1]
    if name not in MODELS:
        print(f"unknown model {name}")
        return 1
    spec = MODELS[name]
    model_path = f"models/{name}/dllama_model_{name}.m"
    tok_path = f"models/{name}/dllama_tokenizer_{name}.t"
    os.makedirs(os.path.dirname(model_path), exist_ok=True)
    ok = all(download(u, model_path) for u in spec["model"]) and \
        all(download(u, tok_path) for u in spec["tokenizer"])
    script = write_run_script(name, model_path, tok_path)
    print(f"📄 wrote {script}" + ("" if ok else " (files still missing)"))
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
