#!/usr/bin/env python3
"""Analyze a hipGraph debug-dot dump (FAA_BENCH_GRAPH_DUMP): for nodes
matching a pattern, print their predecessor/successor kernel labels — a
missing edge between colsum and its dy producer/consumer would localize
the replay corruption to runtime edge derivation.

  python tools/graph_edges.py graph.dot colsum
"""
import re
import sys


def main():
    path = sys.argv[1]
    pat = sys.argv[2] if len(sys.argv) > 2 else "colsum"
    labels = {}
    succs = {}
    preds = {}
    edge_re = re.compile(r'^\s*"?([\w]+)"?\s*->\s*"?([\w]+)"?')
    node_re = re.compile(r'^\s*"?([\w]+)"?\s*\[(.*)\]')
    with open(path) as f:
        for line in f:
            m = edge_re.match(line)
            if m and "->" in line:
                a, b = m.group(1), m.group(2)
                succs.setdefault(a, []).append(b)
                preds.setdefault(b, []).append(a)
                continue
            m = node_re.match(line)
            if m:
                nid, attrs = m.group(1), m.group(2)
                lm = re.search(r'label\s*=\s*"((?:[^"\\]|\\.)*)"', attrs)
                labels[nid] = (lm.group(1)[:200] if lm else attrs[:120])
    print(f"{len(labels)} nodes, {sum(len(v) for v in succs.values())} edges")

    def short(nid):
        l = labels.get(nid, "?").replace("\\n", " ")
        return f"{nid}:{l[:90]}"

    hits = [n for n, l in labels.items() if re.search(pat, l, re.I)]
    print(f"{len(hits)} nodes match '{pat}'")
    for n in hits[:40]:
        print("NODE", short(n))
        for p in preds.get(n, []):
            print("   <-", short(p))
        for s in succs.get(n, []):
            print("   ->", short(s))
    # also report nodes with no predecessors (graph roots) among ALL kernels
    roots = [n for n in labels if n not in preds]
    print(f"{len(roots)} root nodes (no predecessors)")
    for n in roots[:15]:
        print("ROOT", short(n))


if __name__ == "__main__":
    main()
