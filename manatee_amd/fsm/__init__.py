"""The cluster state machine (re-implementation of the external
``manatee-state-machine`` dependency; contract at SURVEY.md §2.2)."""
