"""Embedding encoder tests (CPU reference path + GPU parity marked gpu)."""
import numpy as np
import pytest
import torch

from runbookai_amd.embedding.encoder import BgeEncoder, EncoderConfig


def small_cfg():
    return EncoderConfig(hidden=128, layers=2, heads=2, intermediate=256, max_seq=128)


class TestEncoderCpu:
    def test_shapes_and_norm(self):
        enc = BgeEncoder(device="cpu", cfg=small_cfg())
        vecs = enc.encode(["redis pool exhausted", "gateway timeout", "x"])
        assert vecs.shape == (3, 128)
        norms = np.linalg.norm(vecs, axis=1)
        assert np.allclose(norms, 1.0, atol=1e-4)

    def test_deterministic(self):
        a = BgeEncoder(device="cpu", cfg=small_cfg()).encode(["hello world"])
        b = BgeEncoder(device="cpu", cfg=small_cfg()).encode(["hello world"])
        assert np.allclose(a, b)

    def test_identical_texts_identical_vectors(self):
        enc = BgeEncoder(device="cpu", cfg=small_cfg())
        v = enc.encode(["same text", "same text", "different entirely"])
        assert np.allclose(v[0], v[1], atol=1e-5)
        assert not np.allclose(v[0], v[2], atol=1e-2)


@pytest.mark.gpu
class TestEncoderGpu:
    def test_gpu_matches_cpu(self):
        cfg = small_cfg()
        cpu = BgeEncoder(device="cpu", cfg=cfg).encode(["redis pool exhausted in checkout"])
        gpu = BgeEncoder(device="cuda:0", cfg=cfg).encode(["redis pool exhausted in checkout"])
        cos = float(np.dot(cpu[0], gpu[0]))
        assert cos > 0.99, f"GPU/CPU embedding cosine {cos}"

    def test_full_size_encoder_runs(self):
        enc = BgeEncoder(device="cuda:0")
        vecs = enc.encode(["some operational text"] * 4)
        assert vecs.shape == (4, 384)
        assert np.isfinite(vecs).all()
