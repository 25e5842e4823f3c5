"""Dataset streaming API (config 5 as a feature)."""

import pytest

from helpers import Stack

pa = pytest.importorskip("pyarrow")


@pytest.mark.parametrize("eager", [True, False])
def test_stream_zst_shards_cpu(tmp_path, eager):
    from demodel_amd.engine.datasets import stream_dataset
    from demodel_amd.testing import synth

    files = synth.write_dataset_shards(str(tmp_path / "ds"), n_shards=2,
                                       frames_per_shard=4,
                                       frame_bytes=64 << 10)
    stack = Stack(tmp_path)
    try:
        stack.origin.add_hf_repo("ds/c4", files)
        batches = list(stream_dataset("ds/c4",
                                      endpoint=stack.origin_base,
                                      workers=2, eager=eager))
        assert len(batches) == 2
        codec = pa.Codec("zstd")
        import json

        for b in batches:
            assert len(b.spans) == 4
            # spot-check first span against a CPU decompress
            name = b.name
            idx = json.loads(open(files[name + ".idx.json"]).read())
            raw = open(files[name], "rb").read()
            fr = idx["frames"][0]
            want = bytes(codec.decompress(
                raw[fr["offset"]:fr["offset"] + fr["compressed"]],
                fr["decompressed"]))
            got = bytes(b.tensors()[0].cpu().numpy().tobytes())
            assert got == want
    finally:
        stack.close()


@pytest.mark.gpu
@pytest.mark.parametrize("eager", [True, False])
def test_stream_parquet_gpu(tmp_path, eager):
    import pyarrow.parquet as pq

    from demodel_amd.engine.datasets import stream_dataset
    from demodel_amd.testing import synth

    files = synth.write_parquet_shards(str(tmp_path / "ds"), n_shards=2,
                                       rows_per_shard=20_000)
    stack = Stack(tmp_path)
    try:
        stack.origin.add_hf_repo("ds/pq", files)
        batches = list(stream_dataset("ds/pq",
                                      endpoint=stack.origin_base,
                                      patterns=("*.parquet",),
                                      workers=2, eager=eager))
        assert len(batches) == 2
        for b in batches:
            assert b.data.is_cuda
            assert b.data.numel() == sum(n for _, n in b.spans)
            assert len(b.spans) > 0
    finally:
        stack.close()
