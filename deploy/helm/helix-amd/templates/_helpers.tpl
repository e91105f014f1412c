{{- define "helix-amd.fullname" -}}
{{ .Release.Name }}-helix-amd
{{- end }}
{{- define "helix-amd.labels" -}}
app.kubernetes.io/name: helix-amd
app.kubernetes.io/instance: {{ .Release.Name }}
{{- end }}
