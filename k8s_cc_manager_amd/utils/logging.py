"""Logging setup (reference: fixed-format stdlib logging,
/root/reference/main.py:54-58; --debug raises levels, main.py:727-729)."""

import logging


def setup_logging(debug: bool = False) -> logging.Logger:
    logging.basicConfig(
        level=logging.DEBUG if debug else logging.INFO,
        format="%(asctime)s %(name)s %(levelname)s %(message)s",
    )
    logger = logging.getLogger("cc-manager-amd")
    if debug:
        logger.setLevel(logging.DEBUG)
        logging.getLogger("k8s_cc_manager_amd").setLevel(logging.DEBUG)
    return logger
