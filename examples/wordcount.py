"""Streaming WordCount (the reference's flagship benchmark shape).

Run:  python examples/wordcount.py            (single worker)
      python -m pathway_amd spawn -n 8 examples/wordcount.py   (8 GPUs)
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import pathway_amd as pw

words = pw.io.fs.read("./data", format="plaintext", mode="streaming")
counts = words.groupby(pw.this.data).reduce(
    word=pw.this.data, count=pw.reducers.count()
)
pw.io.csv.write(counts, "word_counts.csv")
pw.run()
