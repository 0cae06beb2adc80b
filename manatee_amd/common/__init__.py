"""Foundations shared by every daemon (ref: lib/common.js, lib/confParser.js)."""
