#!/bin/bash
# Record the same command on several nodes over ssh, one logdir per node in
# the <base>-<ip> convention `sofa report --cluster_ip` merges.
#
# Usage: tools/cluster_record.sh "ip1,ip2,..." "<command>" [logdir-base]
# Assumes passwordless ssh and the repo present at the same path everywhere.
set -e
IPS="$1"; CMD="$2"; BASE="${3:-sofalog}"
REPO="$(cd "$(dirname "$0")/.." && pwd)"
[ -z "$IPS" ] || [ -z "$CMD" ] && { echo "usage: $0 ip1,ip2 \"cmd\" [base]"; exit 2; }
for ip in ${IPS//,/ }; do
  echo "== $ip"
  ssh "$ip" "cd $REPO && python3 bin/sofa stat \"$CMD\" --logdir $BASE-$ip" &
done
wait
echo "merge with: python3 $REPO/bin/sofa report --logdir $BASE --cluster_ip $IPS --skip_preprocess"
