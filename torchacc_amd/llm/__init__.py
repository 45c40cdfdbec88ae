from .qwen_patch import patch_qwen_model  # noqa: F401
