"""Python stack tracing for hang diagnosis.

Reference behavior (xpu_timer/py_xpu_timer/dump_driver.py + stack_viewer.py):
when training hangs, dump the Python stacks of every worker and aggregate
identical stacks across ranks so the operator sees "ranks 0,2,3 stuck in
all_reduce; rank 1 stuck in dataloader" at a glance. The reference drives
gdb/py-spy from a hosting service; the MI355X-native build keeps it
dependency-free: workers self-register a ``faulthandler`` handler on SIGUSR2
(async-signal-safe, in-process, no ptrace needed inside containers), the
elastic agent signals the worker PIDs when the diagnostician requests a
restart for a hang, and the aggregator groups ranks by identical main-thread
stacks.

Worker side is auto-installed by ``import dlrover_amd`` when
``DLROVER_PY_TRACER_DIR`` is set (the agent exports it for its workers).
"""

import os
import re
import signal
import time
from typing import Dict, Iterable, List, Optional

from dlrover_amd.common.log import logger

ENV_TRACER_DIR = "DLROVER_PY_TRACER_DIR"
_installed_fd = None


def stack_path(dump_dir: str, rank: int, pid: int) -> str:
    return os.path.join(dump_dir, f"py_stacks_rank{rank}_pid{pid}.txt")


def install(dump_dir: Optional[str] = None, rank: Optional[int] = None) -> str:
    """Register a SIGUSR2 handler that appends all-thread Python stacks to a
    per-rank file. Returns the file path. Idempotent per process."""
    import faulthandler

    global _installed_fd
    dump_dir = dump_dir or os.getenv(ENV_TRACER_DIR) or "/tmp/dlrover_py_tracer"
    rank = rank if rank is not None else int(os.getenv("RANK", "0"))
    os.makedirs(dump_dir, exist_ok=True)
    path = stack_path(dump_dir, rank, os.getpid())
    if _installed_fd is None:
        # the fd stays open for the process lifetime: faulthandler writes
        # into it directly from the signal handler
        _installed_fd = open(path, "a")
        faulthandler.register(signal.SIGUSR2, file=_installed_fd,
                              all_threads=True, chain=False)
    return path


def maybe_install_from_env() -> Optional[str]:
    if os.getenv(ENV_TRACER_DIR):
        try:
            return install()
        except Exception as e:  # noqa: BLE001 — tracing must never kill a worker
            logger.warning("py_tracer install failed: %s", e)
    return None


def dump_worker_stacks(
    pids: Dict[int, int], dump_dir: str, timeout: float = 3.0
) -> Dict[int, str]:
    """Agent side: SIGUSR2 each worker pid ({local_rank: pid}) and collect the
    stack text each produced. Ranks whose process is gone or silent map to an
    explanatory placeholder instead of raising."""
    sizes = {}
    for lr, pid in pids.items():
        p = stack_path(dump_dir, lr, pid)
        sizes[lr] = os.path.getsize(p) if os.path.exists(p) else 0
        try:
            os.kill(pid, signal.SIGUSR2)
        except OSError as e:
            sizes[lr] = None
            logger.warning("py_tracer: cannot signal pid %s: %s", pid, e)
    deadline = time.time() + timeout
    out: Dict[int, str] = {}
    while time.time() < deadline and len(out) < len(pids):
        for lr, pid in pids.items():
            if lr in out:
                continue
            if sizes[lr] is None:
                out[lr] = "<process not signalable (exited?)>"
                continue
            p = stack_path(dump_dir, lr, pid)
            if os.path.exists(p) and os.path.getsize(p) > sizes[lr]:
                with open(p, errors="replace") as f:
                    f.seek(sizes[lr])
                    out[lr] = f.read()
        time.sleep(0.05)
    for lr in pids:
        out.setdefault(lr, "<no stack produced within timeout>")
    return out


_THREAD_RE = re.compile(r"^(Current thread|Thread) 0x[0-9a-f]+")


def main_thread_stack(text: str) -> List[str]:
    """Extract the 'Current thread' (or first) stack block as a list of
    frame lines, innermost first (faulthandler prints most-recent-first)."""
    blocks: List[List[str]] = []
    cur: Optional[List[str]] = None
    current_idx = None
    for line in text.splitlines():
        if _THREAD_RE.match(line.strip()):
            cur = []
            blocks.append(cur)
            if line.strip().startswith("Current thread"):
                current_idx = len(blocks) - 1
        elif cur is not None and line.strip().startswith("File "):
            cur.append(line.strip())
    if not blocks:
        return []
    return blocks[current_idx if current_idx is not None else 0]


def aggregate_stacks(rank_stacks: Dict[int, str]) -> str:
    """Group ranks by identical main-thread stack (the reference's StackTrie
    leader view, xpu_timer stack_viewer.py:31): one section per distinct
    stack, headed by the rank list holding it."""
    groups: Dict[str, List[int]] = {}
    bodies: Dict[str, List[str]] = {}
    for rank, text in sorted(rank_stacks.items()):
        frames = main_thread_stack(text)
        key = "\n".join(frames) if frames else text.strip()[:2000]
        groups.setdefault(key, []).append(rank)
        bodies[key] = frames or [text.strip()[:2000] or "<empty>"]
    parts = []
    for key, ranks in sorted(groups.items(), key=lambda kv: kv[1]):
        parts.append(
            f"ranks {_fmt_ranks(ranks)} ({len(ranks)} rank(s)):\n  "
            + "\n  ".join(bodies[key])
        )
    return "\n\n".join(parts)


def _fmt_ranks(ranks: Iterable[int]) -> str:
    """Compress [0,1,2,5] -> "0-2,5"."""
    rs = sorted(ranks)
    spans, start, prev = [], rs[0], rs[0]
    for r in rs[1:]:
        if r == prev + 1:
            prev = r
            continue
        spans.append((start, prev))
        start = prev = r
    spans.append((start, prev))
    return ",".join(f"{a}-{b}" if a != b else f"{a}" for a, b in spans)


def read_kernel_stacks(pids: Dict[int, int], max_threads: int = 16) -> Dict[int, str]:
    """Kernel-side stacks of worker processes (/proc/<pid>/task/*/stack,
    root-readable). The MI355X-relevant signal: a rank wedged inside an
    amdgpu wait ioctl shows up here even when its Python stack looks idle —
    the native complement to the SIGUSR2 Python dumps (ref: the reference's
    daemon-side native unwind, stack_util.cc)."""
    import glob
    import os

    out: Dict[int, str] = {}
    for rank, pid in pids.items():
        frames = []
        tasks = sorted(glob.glob(f"/proc/{pid}/task/*/stack"))[:max_threads]
        for t in tasks:
            tid = t.split("/")[-2]
            try:
                with open(t) as f:
                    body = f.read().strip()
            except OSError:
                continue
            if body:
                frames.append(f"  tid {tid}:\n" + "\n".join(
                    "    " + ln for ln in body.splitlines()[:12]))
        if frames:
            out[rank] = "\n".join(frames)
        elif os.path.exists(f"/proc/{pid}"):
            out[rank] = "  (kernel stacks unreadable)"
    return out
