from .jsonutil import js_stringify, js_parse, to_fixed

__all__ = ["js_stringify", "js_parse", "to_fixed"]
