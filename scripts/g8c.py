import sys
from pathlib import Path
sys.path.insert(0, "/root/repo")
sys.path.insert(0, "/root/repo/scripts")
from perf import bench_decode
bench_decode(bs=16, kv=1024, Hq=64, Hkv=8)   # the BASELINE.md decode config
bench_decode(bs=64, kv=1024, Hq=64, Hkv=8)
