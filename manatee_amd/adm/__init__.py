"""Admin library + manatee-adm CLI (ref: lib/adm.js, bin/manatee-adm)."""
