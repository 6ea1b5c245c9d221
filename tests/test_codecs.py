import pytest
import torch

from comfyui_distributed_amd.utils import audio, image


def test_png_roundtrip_exact_uint8():
    t = torch.rand(1, 17, 23, 3)
    b64 = image.encode_png_base64(t)
    back = image.decode_png_base64(b64)
    assert back.shape == (1, 17, 23, 3)
    # PNG is lossless over the uint8 quantization
    q = torch.from_numpy(image.tensor_to_uint8(t[0]))
    q2 = torch.from_numpy(image.tensor_to_uint8(back[0]))
    assert torch.equal(q, q2)


def test_png_bytes_roundtrip():
    t = torch.rand(1, 8, 8, 3)
    raw = image.encode_png_bytes(t)
    back = image.decode_png_bytes(raw)
    assert back.shape == (1, 8, 8, 3)


def test_tensor_to_pil_rejects_batch():
    with pytest.raises(ValueError):
        image.tensor_to_pil(torch.rand(2, 8, 8, 3))


def test_audio_roundtrip():
    a = {"waveform": torch.randn(1, 2, 1000), "sample_rate": 44100}
    env = audio.encode_audio_payload(a)
    assert env["dtype"] == "float32" and env["shape"] == [1, 2, 1000]
    back = audio.decode_audio_payload(env)
    assert torch.allclose(back["waveform"], a["waveform"])
    assert back["sample_rate"] == 44100


def test_audio_validation_errors():
    with pytest.raises(audio.AudioPayloadError):
        audio.encode_audio_payload({"waveform": torch.randn(2, 100)})
    env = audio.encode_audio_payload(
        {"waveform": torch.randn(1, 2, 10), "sample_rate": 8000}
    )
    bad = dict(env)
    bad["shape"] = [1, 2, 11]  # byte-count mismatch
    with pytest.raises(audio.AudioPayloadError):
        audio.decode_audio_payload(bad)
    bad = dict(env)
    bad["dtype"] = "float64"
    with pytest.raises(audio.AudioPayloadError):
        audio.decode_audio_payload(bad)


def test_audio_concat_along_samples():
    a = {"waveform": torch.randn(1, 2, 10), "sample_rate": 8000}
    b = {"waveform": torch.randn(1, 2, 15), "sample_rate": 8000}
    out = audio.concat_audio([a, b])
    assert out["waveform"].shape == (1, 2, 25)
    with pytest.raises(audio.AudioPayloadError):
        audio.concat_audio([a, {"waveform": torch.randn(1, 2, 5), "sample_rate": 4000}])


def test_save_audio_writes_valid_wav(tmp_path):
    import wave

    from comfyui_distributed_amd.graph.builtin_nodes import SaveAudio

    node = SaveAudio()
    node.set_context({"output_dir": str(tmp_path), "saved_images": []})
    t = torch.arange(800, dtype=torch.float32) / 8000.0
    audio = {"waveform": (0.5 * torch.sin(2 * 3.14159 * 440 * t))
             .expand(2, -1).unsqueeze(0).contiguous(),
             "sample_rate": 8000}
    node.save(audio, filename_prefix="tone")
    path = tmp_path / "tone_00000.wav"
    assert path.exists()
    with wave.open(str(path), "rb") as w:
        assert w.getnchannels() == 2
        assert w.getframerate() == 8000
        assert w.getnframes() == 800
        assert w.getsampwidth() == 2


def test_load_audio_roundtrip_through_save(tmp_path):
    from comfyui_distributed_amd.graph.builtin_nodes import LoadAudio, SaveAudio

    ctx = {"output_dir": str(tmp_path), "input_dir": str(tmp_path),
           "saved_images": []}
    la, sa = LoadAudio(), SaveAudio()
    la.set_context(ctx); sa.set_context(ctx)
    src = la.load("synthetic:0.01@8000")[0]
    assert src["waveform"].shape == (1, 2, 80)
    sa.save(src, filename_prefix="rt")
    back = la.load("rt_00000.wav")[0]
    assert back["sample_rate"] == 8000
    assert back["waveform"].shape == (1, 2, 80)
    assert torch.allclose(back["waveform"], src["waveform"], atol=1e-4)
