from tensorlink_amd.engine.formatter import (  # noqa: F401
    ResponseFormatter, extract_reasoning_and_answer, format_chat_prompt,
    normalize_generate_args)
