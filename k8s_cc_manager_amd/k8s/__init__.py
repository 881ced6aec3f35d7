from .client import ApiError, K8sClient, load_client  # noqa: F401
