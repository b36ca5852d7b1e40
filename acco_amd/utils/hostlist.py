"""SLURM hostlist expansion.

Same capability as the reference's utils/hostli.py:9-83 (expand
``n[9-11],gpu-node1`` into ``["n9","n10","n11","gpu-node1"]``) with a compact
from-scratch parser. Used to derive MASTER_ADDR when launched under SLURM.
"""

from __future__ import annotations

import re
from typing import List


class BadHostlist(ValueError):
    pass


def expand_hostlist(hostlist: str) -> List[str]:
    """Expand a SLURM hostlist expression into the list of hostnames."""
    results: List[str] = []
    for part in _split_toplevel(hostlist):
        results.extend(_expand_part(part))
    return results


def _split_toplevel(s: str) -> List[str]:
    """Split on commas that are not inside brackets."""
    parts, depth, cur = [], 0, []
    for ch in s:
        if ch == "[":
            depth += 1
            cur.append(ch)
        elif ch == "]":
            depth -= 1
            if depth < 0:
                raise BadHostlist(f"unbalanced brackets in {s!r}")
            cur.append(ch)
        elif ch == "," and depth == 0:
            if cur:
                parts.append("".join(cur))
                cur = []
        else:
            cur.append(ch)
    if depth != 0:
        raise BadHostlist(f"unbalanced brackets in {s!r}")
    if cur:
        parts.append("".join(cur))
    return parts


def _expand_part(part: str) -> List[str]:
    m = re.match(r"^([^\[]*)\[([^\]]*)\](.*)$", part)
    if not m:
        return [part] if part else []
    prefix, ranges, suffix = m.groups()
    hosts: List[str] = []
    for r in ranges.split(","):
        if "-" in r:
            lo, hi = r.split("-", 1)
            width = len(lo) if lo.startswith("0") else 0
            for i in range(int(lo), int(hi) + 1):
                num = str(i).zfill(width) if width else str(i)
                hosts.extend(_expand_part(prefix + num + suffix))
        else:
            hosts.extend(_expand_part(prefix + r + suffix))
    return hosts


def parse_slurm_tasks_per_node(expr: str) -> List[int]:
    """Expand e.g. ``"2(x3),1"`` into ``[2, 2, 2, 1]``
    (reference utils/hostli.py:317-335 capability)."""
    out: List[int] = []
    for tok in expr.split(","):
        m = re.match(r"^(\d+)(?:\(x(\d+)\))?$", tok)
        if not m:
            raise BadHostlist(f"bad tasks-per-node token {tok!r}")
        n, rep = int(m.group(1)), int(m.group(2) or 1)
        out.extend([n] * rep)
    return out
