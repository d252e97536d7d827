#!/bin/sh
echo "$GREETING from an image-rooted cell"
sleep 3600
