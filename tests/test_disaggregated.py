"""The disaggregated prefill→store→decode demo as a correctness test:
decode-side logits, rebuilt purely from cached KV pages, must match a
monolithic forward pass (CPU here via the TCP fabric; the GPU/local-path
variant lives in test_gpu.py)."""

from infinistore_amd.example.disaggregated import main as disagg_main


def test_disaggregated_decode_cpu(cpu_server):
    disagg_main(port=cpu_server, device="cpu")
