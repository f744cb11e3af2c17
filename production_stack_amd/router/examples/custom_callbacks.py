"""Example custom-callbacks module (reference services/callbacks_service/
custom_callbacks.py pattern). Load with --callbacks
production_stack_amd.router.examples.custom_callbacks."""

import logging

logger = logging.getLogger("router.custom_callbacks")


def pre_request(request, request_json, model):
    """Runs before routing; return a new body dict to rewrite the request,
    or None to leave it unchanged."""
    if request_json.get("max_tokens", 0) > 4096:
        body = dict(request_json)
        body["max_tokens"] = 4096
        logger.info("capped max_tokens for model %s", model)
        return body
    return None


def post_request(request, response):
    return None
