import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import os, torch
os.environ["PW_DEVICE"] = "cuda:0"
import pathway_amd as pw
from pathway_amd.debug import table_from_markdown as T, table_to_dicts

md_lines = ["g | v | __time__ | __diff__"]
import random
rng = random.Random(3)
live = []
serial = 0
for step in range(5):
    et = 2 * (step + 1)
    for _ in range(6):
        serial += 1
        g = rng.choice(["a", "b", "c"])
        v = rng.randint(1, 9)
        md_lines.append(f"{g} | {v} | {et} | 1")
        live.append((g, v, serial))
md = "\n".join(md_lines)

def run():
    pw.internals.rungraph.G.clear()
    t = T(md)
    r = t.groupby(pw.this.g).reduce(pw.this.g, s=pw.reducers.sum(pw.this.v), n=pw.reducers.count())
    _, cols = table_to_dicts(r)
    return sorted(zip(cols["g"].values(), cols["s"].values(), cols["n"].values()))

on = run()
os.environ["PW_NO_SEGRED"] = "1"
import pathway_amd.engine.nodes as N
N._PW_NO_SEGRED = True
off = run()
print("ON :", on)
print("OFF:", off)
print("MATCH" if on == off else "MISMATCH")
