# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/embeddings/bulk_embeddings.py"]
# ---
# # Bulk text embeddings
#
# The fleet-embedding shape: chunk a corpus, `spawn` every chunk (durable,
# pollable), gather embeddings.  The encoder is a small GPT-class transformer
# mean-pooled over tokens, batched per call.

import modal_examples_amd as modal

app = modal.App("example-bulk-embeddings")


@app.cls(gpu="mi355x", scaledown_window=60)
@modal.concurrent(max_inputs=4)
class Embedder:
    @modal.enter()
    def load(self):
        import torch

        from modal_examples_amd.models.gpt.model import GPT, GPTConfig

        self.device = "cuda" if torch.cuda.is_available() else "cpu"
        torch.manual_seed(0)
        self.model = GPT(GPTConfig(n_layer=2, n_embd=128, n_head=2,
                                   block_size=64)).to(self.device)
        if self.device == "cuda":
            self.model = self.model.to(torch.bfloat16)

    @modal.method()
    def embed(self, texts: list) -> list:
        import torch

        ids = torch.zeros(len(texts), 64, dtype=torch.long, device=self.device)
        for i, t in enumerate(texts):
            b = t.encode()[:64]
            ids[i, : len(b)] = torch.tensor(list(b))
        with torch.no_grad():
            logits, _ = self.model(ids)
        emb = logits.float().mean(dim=1)
        emb = emb / emb.norm(dim=-1, keepdim=True)
        return emb.cpu().tolist()


@app.local_entrypoint()
def main(n_docs: int = 64, chunk: int = 16):
    docs = [f"document number {i} about topic {i % 7}" for i in range(n_docs)]
    chunks = [docs[i:i + chunk] for i in range(0, len(docs), chunk)]
    embedder = Embedder()
    calls = [embedder.embed.spawn(c) for c in chunks]
    embs = [e for call in calls for e in call.get()]
    print(f"embedded {len(embs)} docs, dim {len(embs[0])}")
    assert len(embs) == n_docs
