#!/bin/bash
# Round-2 GPU call 12: incremental segmented verification — correctness
# (segmented tests + digest equality vs re-pull) and A/B timing.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

timeout 900 python -m pytest tests/test_resume.py tests/test_gpu_pipeline.py \
    tests/test_gpu_kernels.py tests/test_peer_verify.py -q -m gpu \
    > gpurun_out/pytest_incver.log 2>&1
echo "rc=$?" | tee -a gpurun_out/pytest_incver.log
tail -3 gpurun_out/pytest_incver.log

# digest-consistency probe: segmented pull digests must equal a
# host-computed record (catches any misindexed chunk)
timeout 420 python - > gpurun_out/incver_check.log 2>&1 <<'PYEOF'
import sys, os, hashlib
sys.path.insert(0, "tests")
sys.path.insert(0, ".")
import demodel_amd.engine.pull as pm
pm.SEGMENT_MIN = 3 << 20   # force many ragged segments
pm.MAX_SEGMENTS = 7        # deliberately unaligned bounds
from helpers import Stack
import tempfile, pathlib
td = pathlib.Path(tempfile.mkdtemp())
stack = Stack(td)
data = os.urandom((29 << 20) + 12345)   # ragged total
p = td / "big.bin"; p.write_bytes(data)
stack.origin.add_hf_repo("org/iv", {"big.bin": str(p)})
res = pm.pull_hf("org/iv", endpoint=stack.origin_base, verify="chunked",
                 workers=4)
f = [x for x in res.files if x.name == "big.bin"][0]
vc = f.blob.verify_chunk
want = b"".join(hashlib.sha256(data[o:o+vc]).digest()
                for o in range(0, len(data), vc))
assert f.blob.digest_blob == want, "digest mismatch vs host record"
got = bytes(f.blob.torch_u8().cpu().numpy().tobytes())
assert got == data
print("incremental segmented digests OK:",
      len(f.blob.digest_blob)//32, "chunks")
stack.close()
PYEOF
tail -2 gpurun_out/incver_check.log

# flagship timing with incremental verify
timeout 420 python bench.py --steps 4 --warmup 1 \
    > gpurun_out/bench_incver.json 2> gpurun_out/bench_incver.log
tail -1 gpurun_out/bench_incver.json
timeout 420 python bench.py --model gguf-8b --steps 4 --warmup 1 \
    > gpurun_out/bench_incver_gguf.json 2> gpurun_out/bench_incver_gguf.log
tail -1 gpurun_out/bench_incver_gguf.json

echo DONE
