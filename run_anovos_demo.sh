#!/usr/bin/env bash
# End-to-end demo (the analog of the reference's run_anovos_demo.sh):
# generates the synthetic income dataset, runs the full income pipeline
# on the local device (MI355X when visible, CPU otherwise), and leaves
# ml_anovos_report.html in report_stats/.
set -euo pipefail
cd "$(dirname "$0")"
python tools/make_income_data.py --rows "${ROWS:-100000}" --out data/income_dataset
bin/anovos-run.sh config/configs_full.yaml
echo "report: $(pwd)/report_stats/ml_anovos_report.html"
