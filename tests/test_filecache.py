from metaflow_amd.client.filecache import FileBlobCache


def test_blob_cache_roundtrip(tmp_path):
    c = FileBlobCache(root=str(tmp_path / "cache"), max_bytes=1000)
    assert c.load_key("ab" * 32) is None
    c.store_key("ab" * 32, b"hello")
    assert c.load_key("ab" * 32) == b"hello"


def test_blob_cache_evicts(tmp_path):
    c = FileBlobCache(root=str(tmp_path / "cache"), max_bytes=300)
    import time

    for i in range(5):
        c.store_key(("%02d" % i) * 32, bytes(100))
        time.sleep(0.02)
    # ~2 oldest evicted to fit 300 bytes
    present = [i for i in range(5) if c.load_key(("%02d" % i) * 32)]
    assert len(present) <= 3
    assert 4 in present  # newest kept
