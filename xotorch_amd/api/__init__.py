from xotorch_amd.api.chatgpt import ChatGPTAPI  # noqa: F401
