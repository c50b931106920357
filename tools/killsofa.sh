#!/bin/bash
# Stop a running sofa recording cleanly.
#
# The reference's killsofa.sh (`ps aux | grep sofa | xargs kill -9`) kills by
# PATTERN, which can take down unrelated processes; here we only signal the
# exact PIDs the recorder wrote into <logdir>/sofa_pids.txt.
LOGDIR="${1:-sofalog}"
PIDFILE="$LOGDIR/sofa_pids.txt"
if [ ! -f "$PIDFILE" ]; then
  echo "no $PIDFILE — is a recording running in $LOGDIR?"
  exit 1
fi
while read -r pid; do
  [ -n "$pid" ] && kill -TERM "$pid" 2>/dev/null && echo "sent TERM to $pid"
done < "$PIDFILE"
