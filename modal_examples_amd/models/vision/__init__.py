"""ViT-class vision encoder (K12) for VLM / pdf-RAG examples."""
from .encoder import ViTConfig, VisionEncoder, maxsim
