from tensorlink_amd.api.schemas import (  # noqa: F401
    ChatCompletionRequest, GenerationRequest, JobRequest,
    ModelStatusResponse)
from tensorlink_amd.api.server import TensorlinkAPI, create_app  # noqa: F401
