from .operator_server import create_operator_app, make_token, parse_token

__all__ = ["create_operator_app", "make_token", "parse_token"]
