"""Dev/operator tooling (ref: tools/mkdevsitters, tools/mksitterconfig,
test/testManatee.js)."""
