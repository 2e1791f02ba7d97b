{{- define "forge.fullname" -}}
{{- printf "%s" .Release.Name | trunc 63 | trimSuffix "-" -}}
{{- end -}}
{{- define "forge.labels" -}}
app.kubernetes.io/name: mcp-context-forge-amd
app.kubernetes.io/instance: {{ .Release.Name }}
{{- end -}}
