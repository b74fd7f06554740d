#!/usr/bin/env bash
# Model preset selector — reference set_llm_config.sh (125M/1B/3B/7B).
# Prints the override to pass to the launch scripts.
case "${1:-125m}" in
  125m) echo "llm_config=mpt-125m" ;;
  350m) echo "llm_config=mpt-350m" ;;
  1b)   echo "llm_config=mpt-1b" ;;
  3b)   echo "llm_config=mpt-3b" ;;
  7b)   echo "llm_config=mpt-7b" ;;
  *) echo "usage: $0 {125m|350m|1b|3b|7b}" >&2; exit 1 ;;
esac
