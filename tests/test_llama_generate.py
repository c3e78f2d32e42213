"""KV-cached decoding correctness: generate() with the per-layer cache
must produce exactly the tokens (and matching logits) of the no-cache
oracle that re-runs the full forward at every step."""
import torch

from axon.models.llama import LlamaForCausalLM, llama_tiny_config


def _greedy_no_cache(model, ids, n):
    for _ in range(n):
        logits = model.forward(ids)[:, -1].float()
        ids = torch.cat([ids, logits.argmax(-1, keepdim=True)], 1)
    return ids


def test_generate_matches_no_cache_oracle():
    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny_config()).eval()
    ids = torch.randint(0, 256, (2, 7))
    with torch.no_grad():
        want = _greedy_no_cache(model, ids.clone(), 9)
    got = model.generate(ids.clone(), 9)
    assert torch.equal(got, want), (got, want)


def test_generate_cached_logits_match_full_forward():
    torch.manual_seed(1)
    model = LlamaForCausalLM(llama_tiny_config()).eval()
    ids = torch.randint(0, 256, (1, 5))
    out = model.generate(ids, 6)
    with torch.no_grad():
        # prefill then single-token steps, comparing each step's logits to
        # a fresh full forward over the same prefix
        caches = [{"k": None, "v": None} for _ in model.blocks]
        full = model.forward(out[:, :-1])
        inc = model.forward(out[:, :5], caches, 0)
        pos = 5
        steps = [inc[:, -1]]
        for t in range(5, out.shape[1] - 1):
            inc = model.forward(out[:, t:t + 1], caches, pos)
            pos += 1
            steps.append(inc[:, -1])
        for i, s in enumerate(steps):
            ref = full[:, 4 + i]
            assert torch.allclose(s, ref, rtol=1e-4, atol=1e-5), i


def test_generate_sampling_shapes():
    torch.manual_seed(2)
    model = LlamaForCausalLM(llama_tiny_config()).eval()
    ids = torch.randint(0, 256, (3, 4))
    out = model.generate(ids, 5, temperature=0.8, top_k=10)
    assert out.shape == (3, 9)
    assert torch.equal(out[:, :4], ids)
    assert int(out.max()) < 256 and int(out.min()) >= 0
