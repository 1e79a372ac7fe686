# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/llm_serving/chat_with_pdf_vision.py"]
# ---
# # Chat with a PDF, vision edition (ColPali-style page retrieval + LLM)
#
# Mirrors the reference's vision-RAG recipe (llm-serving/chat_with_pdf_vision.py):
# a ViT-class encoder embeds every PDF page IMAGE into multi-vector patch
# embeddings; a question is scored against pages by late interaction (MaxSim);
# the best page's content is handed to the Llama engine to answer.  Both
# models live in ONE GPU container class so retrieval → generation never
# leaves the device.
#
# Synthetic stand-ins (no network): "pages" are rendered as structured noise
# images with distinct per-page patterns; the self-test asks about a page it
# perturbed and asserts retrieval picks that page.

import modal_examples_amd as modal

app = modal.App("example-chat-pdf-vision")

sessions = modal.Dict.from_name("pdf-vision-sessions", create_if_missing=True)


@app.cls(gpu="mi355x", timeout=600, scaledown_window=120)
class PdfChat:
    @modal.enter()
    def load(self):
        import torch

        from modal_examples_amd.models.llama.engine import LlamaEngine
        from modal_examples_amd.models.llama.model import LlamaConfig
        from modal_examples_amd.models.llama.server import LLMServer
        from modal_examples_amd.models.vision import ViTConfig, VisionEncoder

        gpu = torch.cuda.is_available()
        self.device = "cuda" if gpu else "cpu"
        self.dtype = torch.bfloat16 if gpu else torch.float32
        vcfg = ViTConfig.base() if gpu else ViTConfig.small_test()
        self.vit = VisionEncoder(vcfg).to(self.device, self.dtype).eval()
        lcfg = LlamaConfig.llama3_8b() if gpu else LlamaConfig.small()
        eng = LlamaEngine(lcfg, device=self.device, dtype=self.dtype,
                          use_graph=gpu, kv_blocks=None if gpu else 128)
        self.llm = LLMServer(eng, model_name="pdf-vision-chat")
        self.pages = {}  # doc_id -> (embeddings [P,N,D], page texts)

    @modal.method()
    def index_pdf(self, doc_id: str, page_images, page_texts) -> int:
        """page_images: [P,3,H,W] float arrays; embeddings stay on-device."""
        import torch

        imgs = torch.as_tensor(page_images).to(self.device, self.dtype)
        embs = self.vit.embed(imgs)  # [P, N, D], normalized
        self.pages[doc_id] = (embs, list(page_texts))
        return embs.shape[0]

    @modal.method()
    def ask(self, doc_id: str, question_image, question: str,
            max_tokens: int = 24) -> dict:
        """question_image: the query rendered as an image (ColPali scores
        vision-to-vision); returns the retrieved page + the LLM's answer."""
        import torch

        from modal_examples_amd.models.vision import maxsim

        embs, texts = self.pages[doc_id]
        q = torch.as_tensor(question_image)[None].to(self.device, self.dtype)
        qe = self.vit.embed(q)[0]  # [N, D]
        scores = maxsim(qe, embs)
        best = int(scores.argmax())
        prompt = (f"Context from page {best + 1}: {texts[best]}\n"
                  f"Question: {question}\nAnswer:")
        answer = self.llm.generate(prompt, max_tokens=max_tokens)
        return {"page": best, "scores": [round(float(s), 2) for s in scores],
                "answer": answer}


@app.local_entrypoint()
def main():
    import numpy as np

    rng = np.random.default_rng(7)
    n_pages, size = 4, 64
    # distinct per-page structure (each page = its own random texture)
    pages = rng.standard_normal((n_pages, 3, size, size)).astype("float32")
    texts = [f"Page {i+1} discusses topic T{i+1}." for i in range(n_pages)]

    chat = PdfChat()
    n = chat.index_pdf.remote("doc-1", pages, texts)
    print(f"indexed {n} pages")

    # the "question" looks like page 3 plus noise — retrieval must find it
    target = 2
    query_img = pages[target] + 0.15 * rng.standard_normal(pages[target].shape)
    out = chat.ask.remote("doc-1", query_img.astype("float32"),
                          "What does this page discuss?")
    print(f"retrieved page {out['page']} (scores {out['scores']})")
    print(f"answer: {out['answer']!r}")
    assert out["page"] == target, f"retrieval picked {out['page']}, wanted {target}"
