"""Run bench.py against a frozen engine .so (A/B regression probe).
Usage: python tools/bench_oldso.py <path-to-so> [bench args...]"""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import cassandra_amd as ca
ca.load_library(sys.argv[1])
sys.argv = ["bench.py"] + sys.argv[2:]
_bench = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "bench.py")
exec(compile(open(_bench).read(), _bench, "exec"), {"__name__": "__main__", "__file__": _bench})
