"""Backup plane: REST job server + snapshot sender + restore client
(ref: lib/backupServer.js, lib/backupQueue.js, lib/backupSender.js,
lib/zfsClient.js restore path)."""
