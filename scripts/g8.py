import sys
from pathlib import Path
sys.path.insert(0, "/root/repo")
sys.path.insert(0, "/root/repo/scripts")
from perf import bench_decode
bench_decode(bs=256, kv=8192, Hq=64, Hkv=8)
bench_decode(bs=256, kv=8192, Hq=32, Hkv=2)  # GROUP=16
