"""Operator CLIs (cf. nvidiagpuplugin/cmd/, nvmlinfo/)."""
