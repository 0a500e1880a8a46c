"""cProfile wrapper for the RAG retrieve bench (diagnostics)."""
import cProfile
import os
import pstats
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from scripts.bench_rag import bench_rag_serving

cProfile.run(
    "bench_rag_serving(n_docs=20000, n_queries=60)", "gpurun_out/rag.prof"
)
s = pstats.Stats("gpurun_out/rag.prof")
s.sort_stats("cumulative").print_stats(28)
