"""Prometheus exposition aggregation (reference: model_gateway/src/worker/
metrics_aggregator.rs — parse each worker's /metrics text, stamp per-worker
labels, merge families across workers with label-set padding, re-emit one
exposition at /engine_metrics)."""
from __future__ import annotations

import re
from typing import Dict, List, Tuple

_SAMPLE_RE = re.compile(r"^([a-zA-Z_][a-zA-Z0-9_]*)(?:\{(.*)\})?\s+(\S+)(?:\s+\d+)?$")
_LABEL_RE = re.compile(r'([a-zA-Z_][a-zA-Z0-9_]*)="((?:[^"\\]|\\.)*)"')


class Family:
    __slots__ = ("name", "kind", "help", "samples")

    def __init__(self, name: str, kind: str = "untyped", help_: str = ""):
        self.name = name
        self.kind = kind
        self.help = help_
        # (sample_name, labels tuple-of-pairs, value string)
        self.samples: List[Tuple[str, Tuple[Tuple[str, str], ...], str]] = []


def parse_prometheus(text: str) -> Dict[str, Family]:
    """Prometheus text format -> families.  Colons in names are replaced with
    underscores (metrics_aggregator.rs:19)."""
    text = text.replace(":", "_")
    families: Dict[str, Family] = {}
    for line in text.splitlines():
        line = line.strip()
        if not line:
            continue
        if line.startswith("# HELP "):
            parts = line.split(None, 3)
            if len(parts) >= 3:
                fam = families.setdefault(parts[2], Family(parts[2]))
                fam.help = parts[3] if len(parts) > 3 else ""
            continue
        if line.startswith("# TYPE "):
            parts = line.split(None, 3)
            if len(parts) >= 4:
                fam = families.setdefault(parts[2], Family(parts[2]))
                fam.kind = parts[3]
            continue
        if line.startswith("#"):
            continue
        m = _SAMPLE_RE.match(line)
        if not m:
            continue
        sname, labels_raw, value = m.group(1), m.group(2), m.group(3)
        # histogram/summary sample names belong to their base family
        base = re.sub(r"_(bucket|sum|count|total)$", "", sname)
        fam = families.get(sname) or families.get(base)
        if fam is None:
            fam = families.setdefault(sname, Family(sname))
        labels = tuple(sorted((k, v) for k, v in _LABEL_RE.findall(labels_raw or "")))
        fam.samples.append((sname, labels, value))
    return families


def _with_labels(fam: Family, extra: List[Tuple[str, str]]) -> Family:
    out = Family(fam.name, fam.kind, fam.help)
    for sname, labels, value in fam.samples:
        out.samples.append((sname, tuple(sorted({**dict(labels), **dict(extra)}.items())), value))
    return out


def _align_labels(families: List[Family]) -> None:
    """Pad missing labels with "" so all samples in a family share one label
    set (metrics_aggregator.rs align_labels)."""
    names = set()
    for fam in families:
        for _, labels, _ in fam.samples:
            names.update(k for k, _ in labels)
    for fam in families:
        padded = []
        for sname, labels, value in fam.samples:
            d = dict(labels)
            for n in names:
                d.setdefault(n, "")
            padded.append((sname, tuple(sorted(d.items())), value))
        fam.samples = padded


def merge_expositions(packs: List[Tuple[str, Dict[str, Family]]]) -> str:
    """[(worker_label, families)] -> one merged exposition text.  Each pack's
    samples get a `worker` label; same-name families merge with label-set
    alignment."""
    merged: Dict[str, List[Family]] = {}
    for worker, families in packs:
        for name, fam in families.items():
            merged.setdefault(name, []).append(_with_labels(fam, [("worker", worker)]))
    lines: List[str] = []
    for name in sorted(merged):
        fams = merged[name]
        _align_labels(fams)
        kind = next((f.kind for f in fams if f.kind != "untyped"), "untyped")
        help_ = next((f.help for f in fams if f.help), "")
        if help_:
            lines.append(f"# HELP {name} {help_}")
        lines.append(f"# TYPE {name} {kind}")
        for fam in fams:
            for sname, labels, value in fam.samples:
                if labels:
                    lbl = ",".join(f'{k}="{v}"' for k, v in labels)
                    lines.append(f"{sname}{{{lbl}}} {value}")
                else:
                    lines.append(f"{sname} {value}")
    return "\n".join(lines) + ("\n" if lines else "")


async def collect_engine_metrics(ctx, session=None) -> str:
    """Fetch every worker's /metrics and merge (WorkerManager::
    get_engine_metrics).  Data-plane (sim://, rccl://) workers synthesize an
    exposition from their tracked counters."""
    packs: List[Tuple[str, Dict[str, Family]]] = []
    for w in ctx.worker_registry.all():
        if w.url.startswith(("http://", "https://")) and session is not None:
            try:
                async with session.get(w.url.rstrip("/") + "/metrics") as resp:
                    if resp.status == 200:
                        packs.append((w.url, parse_prometheus(await resp.text())))
                        continue
            except Exception:
                continue
        synth = (
            f"# TYPE smg_worker_active_requests gauge\nsmg_worker_active_requests {w.active_requests}\n"
            f"# TYPE smg_worker_token_usage gauge\nsmg_worker_token_usage {w.token_usage}\n"
            f"# TYPE smg_worker_gen_throughput gauge\nsmg_worker_gen_throughput {w.gen_throughput}\n"
            f"# TYPE smg_worker_processed_total counter\nsmg_worker_processed_total {w.processed_requests}\n"
        )
        packs.append((w.url, parse_prometheus(synth)))
    return merge_expositions(packs)
