"""HF-hub model resolution (reference: crates/tokenizer/src/hub.rs —
model-id -> local cache dir via the HF cache layout, with download when the
hub is reachable).

Resolution order for `resolve_model_dir(id_or_path)`:
  1. an existing local path (dir or file) is returned as-is;
  2. the HF cache layout ($HF_HOME|~/.cache/huggingface)/hub/
     models--{org}--{name}/snapshots/<rev>/ — newest snapshot containing
     tokenizer artifacts wins (hub.rs:232 resolve_model_cache_dir);
  3. huggingface_hub.snapshot_download — local_files_only unless
     allow_download=True (this image has no egress; download is the
     reference behavior when a hub IS reachable).
"""
from __future__ import annotations

import os
from typing import List, Optional

TOKENIZER_FILES = (
    "tokenizer.json",
    "tokenizer_config.json",
    "vocab.json",
    "merges.txt",
    "tokenizer.model",
)


def hf_cache_home() -> str:
    if os.environ.get("HF_HUB_CACHE"):
        return os.environ["HF_HUB_CACHE"]
    base = os.environ.get("HF_HOME") or os.path.join(
        os.environ.get("XDG_CACHE_HOME", os.path.expanduser("~/.cache")), "huggingface")
    return os.path.join(base, "hub")


def _has_tokenizer_files(d: str) -> bool:
    try:
        names = set(os.listdir(d))
    except OSError:
        return False
    return any(f in names for f in TOKENIZER_FILES) or any(
        n.endswith(".tiktoken") for n in names)


def _snapshot_dirs(model_id: str) -> List[str]:
    repo_dir = os.path.join(hf_cache_home(), "models--" + model_id.replace("/", "--"))
    snaps = os.path.join(repo_dir, "snapshots")
    if not os.path.isdir(snaps):
        return []
    out = []
    for rev in os.listdir(snaps):
        p = os.path.join(snaps, rev)
        if os.path.isdir(p):
            out.append(p)
    # newest snapshot first (mtime); refs/main would be more precise but a
    # cache can lack refs — mtime matches hub.rs's practical fallback
    out.sort(key=lambda p: os.path.getmtime(p), reverse=True)
    return out


def resolve_model_dir(id_or_path: str, allow_download: bool = False) -> Optional[str]:
    """Model id or path -> a local directory with tokenizer artifacts."""
    if os.path.exists(id_or_path):
        return id_or_path
    for snap in _snapshot_dirs(id_or_path):
        if _has_tokenizer_files(snap):
            return snap
    # hub fallback (no egress in this image unless the operator enables it)
    try:
        from huggingface_hub import snapshot_download

        return snapshot_download(
            id_or_path,
            allow_patterns=["tokenizer*", "*.tiktoken", "vocab*", "merges*",
                            "special_tokens_map.json", "*.model"],
            local_files_only=not allow_download,
        )
    except Exception:
        return None


def load_tokenizer(id_or_path: str, name: Optional[str] = None,
                   allow_download: bool = False):
    """Resolve + construct the right tokenizer family: tiktoken-format dirs
    get TiktokenTokenizer (incl. Kimi-K2 pattern detection), everything else
    the HF tokenizer.json loader."""
    d = resolve_model_dir(id_or_path, allow_download=allow_download)
    if d is None:
        raise FileNotFoundError(
            f"cannot resolve tokenizer {id_or_path!r} (not a path, not in the "
            f"HF cache at {hf_cache_home()}, hub unreachable)")
    from .tiktoken_bpe import TiktokenTokenizer, is_tiktoken_dir

    if os.path.isdir(d) and is_tiktoken_dir(d):
        return TiktokenTokenizer.from_dir(d, name=name)
    if os.path.isfile(d) and d.endswith(".tiktoken"):
        return TiktokenTokenizer.from_file(d, name=name)
    from .registry import HFTokenizer

    return HFTokenizer(d, name)
